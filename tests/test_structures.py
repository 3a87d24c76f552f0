"""Batched structures + TensorFrame tests (mirrors reference
tests/test_structures.py and test_tensorframe.py)."""

import pytest
import torch

from evotorch_amd.utils import CBag, CDict, CList, CMemory, TensorFrame


def test_cmemory_basic():
    m = CMemory(3, num_keys=5, fill_with=0.0)
    m.set_(2, torch.tensor([1.0, 2.0, 3.0]))
    assert torch.allclose(m[2], torch.tensor([1.0, 2.0, 3.0]))
    m.add_(2, 1.0)
    assert torch.allclose(m[2], torch.tensor([2.0, 3.0, 4.0]))
    assert torch.allclose(m[0], torch.zeros(3))


def test_cmemory_batched_masked():
    m = CMemory(num_keys=4, batch_size=3, fill_with=0.0)
    keys = torch.tensor([0, 1, 2])
    m.set_(keys, torch.tensor([10.0, 20.0, 30.0]))
    got = m.get(keys)
    assert torch.allclose(got, torch.tensor([10.0, 20.0, 30.0]))
    # masked write: only batch elements 0 and 2
    m.set_(keys, torch.tensor([-1.0, -2.0, -3.0]), where=torch.tensor([True, False, True]))
    assert torch.allclose(m.get(keys), torch.tensor([-1.0, 20.0, -3.0]))


def test_cmemory_tuple_keys():
    m = CMemory(num_keys=(3, 5), fill_with=0.0)
    m.set_(torch.tensor([2, 4]), 7.0)
    assert float(m.get(torch.tensor([2, 4]))) == 7.0
    with pytest.raises(KeyError):
        m.get(torch.tensor([3, 0]))


def test_cmemory_key_offset():
    m = CMemory(num_keys=10, key_offset=1)
    m.set_(1, 5.0)
    assert float(m[1]) == 5.0
    with pytest.raises(KeyError):
        m.get(0)


def test_cdict_presence_and_default():
    d = CDict(num_keys=6, batch_size=2, fill_with=0.0)
    keys = torch.tensor([1, 3])
    assert not bool(d.contains(keys).any())
    d.set_(keys, torch.tensor([4.0, 5.0]))
    assert bool(d.contains(keys).all())
    other = torch.tensor([2, 2])
    got = d.get(other, default=-1.0)
    assert torch.allclose(got, torch.tensor([-1.0, -1.0]))
    d.clear()
    assert not bool(d.contains(keys).any())


def test_clist_push_pop():
    lst = CList(max_length=4, batch_size=2)
    lst.append_(torch.tensor([1.0, 10.0]))
    lst.append_(torch.tensor([2.0, 20.0]))
    assert lst.length.tolist() == [2, 2]
    assert torch.allclose(lst[0], torch.tensor([1.0, 10.0]))
    assert torch.allclose(lst[-1], torch.tensor([2.0, 20.0]))
    popped = lst.pop_()
    assert torch.allclose(popped, torch.tensor([2.0, 20.0]))
    assert lst.length.tolist() == [1, 1]
    lst.appendleft_(torch.tensor([0.0, 0.5]))
    assert torch.allclose(lst[0], torch.tensor([0.0, 0.5]))
    left = lst.popleft_()
    assert torch.allclose(left, torch.tensor([0.0, 0.5]))


def test_clist_masked_append():
    lst = CList(max_length=3, batch_size=2)
    lst.append_(torch.tensor([1.0, 1.0]))
    lst.append_(torch.tensor([2.0, 2.0]), where=torch.tensor([True, False]))
    assert lst.length.tolist() == [2, 1]


def test_clist_capacity():
    lst = CList(max_length=2, batch_size=1)
    for v in [1.0, 2.0, 3.0]:
        lst.append_(torch.tensor([v]))
    assert lst.length.tolist() == [2]  # third append ignored (full)


def test_cbag_pop_returns_pushed_values():
    g = torch.Generator().manual_seed(0)
    bag = CBag(max_length=5, batch_size=1, generator=g)
    for v in [1.0, 2.0, 3.0]:
        bag.push_(torch.tensor([v]))
    seen = set()
    for _ in range(3):
        seen.add(float(bag.pop_()[0]))
    assert seen == {1.0, 2.0, 3.0}
    assert bag.length.tolist() == [0]


# -- TensorFrame -------------------------------------------------------------


def test_tensorframe_basics():
    tf = TensorFrame({"a": torch.arange(5.0), "b": torch.arange(5.0) * 2})
    assert len(tf) == 5
    assert tf.columns == ["a", "b"]
    assert torch.allclose(tf["b"], torch.arange(5.0) * 2)
    tf["c"] = torch.ones(5)
    assert "c" in tf


def test_tensorframe_sort_and_pick():
    tf = TensorFrame({"x": torch.tensor([3.0, 1.0, 2.0]), "y": torch.tensor([30.0, 10.0, 20.0])})
    s = tf.sort("x")
    assert s["y"].tolist() == [10.0, 20.0, 30.0]
    sub = tf.pick[torch.tensor([0, 2])]
    assert sub["x"].tolist() == [3.0, 2.0]
    sub2 = tf.pick[torch.tensor([True, False, True])]
    assert sub2["y"].tolist() == [30.0, 20.0]
    one_col = tf.pick[torch.tensor([0, 1]), "y"]
    assert one_col.columns == ["y"]


def test_tensorframe_stack():
    a = TensorFrame({"x": torch.zeros(2)})
    b = TensorFrame({"y": torch.ones(2)})
    h = a.hstack(b)
    assert set(h.columns) == {"x", "y"}
    c = TensorFrame({"x": torch.ones(3)})
    v = a.vstack(c)
    assert len(v) == 5
    with pytest.raises(ValueError):
        a.vstack(b)


def test_tensorframe_each():
    tf = TensorFrame({"x": torch.arange(4.0), "y": torch.ones(4)})
    out = tf.each(lambda row: {"z": row["x"] + row["y"]})
    assert out["z"].tolist() == [1.0, 2.0, 3.0, 4.0]
    joined = tf.each(lambda row: {"z": row["x"] * 2}, join=True)
    assert set(joined.columns) == {"x", "y", "z"}


def test_tensorframe_read_only():
    tf = TensorFrame({"a": torch.zeros(3)}).get_read_only_view()
    with pytest.raises(TypeError):  # reference raises TypeError
        tf["b"] = torch.ones(3)


def test_tensorframe_to_pandas():
    tf = TensorFrame({"a": torch.arange(3.0)})
    df = tf.to_pandas()
    assert list(df["a"]) == [0.0, 1.0, 2.0]


def test_tensorframe_pandas_style_methods():
    tf = TensorFrame({"a": torch.tensor([3.0, 1.0, 2.0]), "b": torch.tensor([30.0, 10.0, 20.0])})
    assert tf.nlargest(2, "a")["a"].tolist() == [3.0, 2.0]
    assert tf.nsmallest(1, "b")["b"].tolist() == [10.0]
    assert tf.drop(columns="b").columns == ["a"]
    joined = tf.drop(columns="b").join(TensorFrame({"c": torch.ones(3)}))
    assert set(joined.columns) == {"a", "c"}
    x = tf.as_tensor(5, to_work_with="a", broadcast_if_scalar=True)
    assert x.shape == (3,) and x.dtype == torch.float32
    assert tf.cpu().device == torch.device("cpu")


# -- reference-parity behavior set (ported from the reference's
# test_tensorframe.py / test_objectarray.py BEHAVIORS, original code) -------


def test_tensorframe_pick_setter_isolated_from_source():
    src = torch.tensor([1.0, 2, 3, 4, 5])
    tf = TensorFrame({"X": src, "Y": src * 10})
    tf.pick[1:4, "X"] = torch.tensor([-2.0, -3, -4])
    assert tf.X.tolist() == [1, -2, -3, -4, 5]
    assert src.tolist() == [1, 2, 3, 4, 5]  # ctor copied; source untouched


@pytest.mark.parametrize("rhs_as_frame", [False, True])
def test_tensorframe_pick_setter_multicolumn(rhs_as_frame):
    src = torch.tensor([1.0, 2, 3, 4, 5])
    tf = TensorFrame({"X": src.clone(), "Y": (src * 10).clone()})
    rhs = {"X": torch.tensor([-2.0, -3, -4]), "Y": torch.tensor([-20.0, -30, -40])}
    tf.pick[1:4, ["X", "Y"]] = TensorFrame(rhs) if rhs_as_frame else rhs
    assert tf.X.tolist() == [1, -2, -3, -4, 5]
    assert tf.Y.tolist() == [10, -20, -30, -40, 50]


def test_tensorframe_each_under_vmap_with_scalar_column():
    """The reference's batched-operations contract: a whole frame pipeline
    (ctor + each with a dict row fn + scalar column) composes under an
    outer torch.func.vmap."""
    torch.manual_seed(0)
    bx = torch.randn(2, 3, 2)
    by = torch.randn(2, 3)

    def run(x, y):
        tf = TensorFrame(dict(X=x, Y=y, Z=True))

        def per_row(row):
            assert row["X"].shape == (2,)
            assert row["Y"].shape == ()
            assert row["Z"].shape == ()
            return {"OUT": row["Z"] * (torch.max(row["X"]) + row["Y"])}

        return tf.each(per_row).OUT

    out = torch.func.vmap(run)(bx, by)
    want = torch.func.vmap(torch.func.vmap(torch.max))(bx) + by
    assert torch.allclose(out, want, atol=1e-5)


def test_tensorframe_pick_column_slicers_and_bool_mask():
    tf = TensorFrame(dict(A=[1.0, 2, 3, 4], B=[[10.0, 20], [30, 40], [50, 60], [70, 80]], C=[-1.0, -2, -3, -4]))
    sub = tf.pick[[1, 3]]
    assert sub.columns == ["A", "B", "C"] and sub.A.tolist() == [2, 4]
    assert sub.B.tolist() == [[30, 40], [70, 80]]
    sub = tf.pick[[1, 3], "A"]
    assert sub.columns == ["A"]
    for slicer in (slice(None, 2), tf.C > -3):
        assert tf.pick[slicer, "A"].A.tolist() == [1, 2]
        got = tf.pick[slicer, ["A", "C"]]
        assert got.columns == ["A", "C"] and got.C.tolist() == [-1, -2]
        full = tf.pick[slicer, slice(None)]
        assert full.columns == ["A", "B", "C"]


def test_tensorframe_vstack_failures():
    a = TensorFrame(dict(A=[1.0, 2, 3], B=[4.0, 5, 6]))
    with pytest.raises(ValueError):
        a.vstack(TensorFrame(dict(A=[1.0, 2, 3], B=[[4.0, 5], [6, 7], [8, 9]])))
    with pytest.raises(ValueError):
        a.vstack(TensorFrame(dict(A=[1.0, 2, 3], C=[1.0, 2, 3])))


def test_tensorframe_read_only_full_protocol():
    tf = TensorFrame(dict(A=[1.0, 2, 3]))
    tf["B"] = 4  # scalar broadcast
    assert tf.B.tolist() == [4, 4, 4]
    ro = tf.get_read_only_view()
    with pytest.raises(TypeError):
        ro["C"] = 5
    with pytest.raises(TypeError):
        ro["A"][:] = 2.0  # columns of a read-only view are read-only
    cloned = ro.clone()
    assert not cloned.is_read_only
    cloned["C"] = 5
    cloned["A"][0] = 9.0
    # storing a frame in an ObjectArray makes it immutable (read-only view)
    from evotorch_amd.utils.objectarray import ObjectArray

    arr = ObjectArray(1)
    arr[0] = cloned
    assert arr[0].is_read_only
    with pytest.raises(TypeError):
        arr[0]["D"] = 10


def test_tensorframe_with_columns_preserves_read_only():
    tf = TensorFrame(dict(A=[1.0, 2, 3])).with_columns(B=[4.0, 5, 6], C=7)
    assert tf.C.tolist() == [7, 7, 7]
    ro = tf.get_read_only_view().with_columns(A=[10.0, 20, 30], Z=[100.0, 200, 300])
    assert ro.is_read_only
    assert ro.A.tolist() == [10, 20, 30] and ro.Z.tolist() == [100, 200, 300]


def test_objectarray_copy_and_deepcopy_protocols():
    from copy import copy, deepcopy

    from evotorch_amd.utils.objectarray import ObjectArray

    x = ObjectArray(10)
    x[:] = [0 for _ in range(10)]
    for maker in (copy, deepcopy, lambda a: a.clone()):
        y = maker(x)
        assert all(a == b for a, b in zip(x, y))
        y[:] = [1 for _ in range(10)]
        assert all(a != b for a, b in zip(x, y))  # independent storage


def test_objectarray_clone_clones_elements():
    from evotorch_amd.utils.objectarray import ObjectArray

    x = ObjectArray(2)
    x[0] = [1, 2]
    x[1] = [3, 4]
    y = x.clone()
    assert x[0] == y[0] and x[1] == y[1]
    assert x[0] is not y[0] and x[1] is not y[0]


def test_objectarray_storage_ptr_shared_by_views():
    from evotorch_amd.utils.misc import storage_ptr
    from evotorch_amd.utils.objectarray import ObjectArray

    x = ObjectArray(10)
    x[:] = range(10)
    y = x[3:5]
    assert storage_ptr(x) == storage_ptr(y)
    y[:] = [0, 0]
    assert x[3] == 0 and x[4] == 0
    ro = x.get_read_only_view()
    assert storage_ptr(ro) == storage_ptr(x)
    with pytest.raises(ValueError):
        ro[0] = 9
