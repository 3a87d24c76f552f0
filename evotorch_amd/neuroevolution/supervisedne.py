"""SupervisedNE: dataset + loss neuroevolution.

Reference parity: /root/reference/src/evotorch/neuroevolution/
supervisedne.py:30-348 — minibatch-based fitness with the
`common_minibatch` low-variance mode (all solutions of a batch are scored
on the SAME minibatch, so their fitnesses are directly comparable).

MI355X-native evaluation path: instead of looping solutions through one
network, the whole population is evaluated on the minibatch in a single
vmapped batched forward (rocBLAS batched GEMMs) when
`vectorized_eval=True` (default for common_minibatch mode).
"""

from typing import Callable, Optional, Union

import torch
from torch import nn
from torch.utils.data import DataLoader, Dataset

from ..core import SolutionBatch
from ..models import make_functional_module
from .neproblem import NEProblem

__all__ = ["SupervisedNE"]


class SupervisedNE(NEProblem):
    def __init__(
        self,
        dataset: Dataset,
        network: Union[str, nn.Module, Callable[[], nn.Module]],
        loss_func: Optional[Callable] = None,
        *,
        network_args: Optional[dict] = None,
        initial_bounds=(-0.00001, 0.00001),
        minibatch_size: Optional[int] = None,
        num_minibatches: Optional[int] = None,
        common_minibatch: bool = True,
        loss_as_fitness_sign: str = "min",
        device=None,
        seed: Optional[int] = None,
        vectorized_eval: bool = True,
        subbatch_size: Optional[int] = None,
        num_subbatches: Optional[int] = None,
        num_actors=None,
        num_gpus_per_actor=None,
        actor_config=None,
    ):
        # num_subbatches/subbatch_size keep their SupervisedNE meaning
        # (solutions per common minibatch) and are consumed here, not by
        # the evaluation pool.
        super().__init__(
            "min" if loss_as_fitness_sign == "min" else "max",
            network,
            network_args=network_args,
            initial_bounds=initial_bounds,
            device=device,
            seed=seed,
            num_actors=num_actors,
            num_gpus_per_actor=num_gpus_per_actor,
            actor_config=actor_config,
        )
        self._dataset = dataset
        self._loss_func = loss_func
        self._minibatch_size = None if minibatch_size is None else int(minibatch_size)
        self._num_minibatches = max(1, int(num_minibatches)) if num_minibatches is not None else 1
        self._common_minibatch = bool(common_minibatch)
        self._vectorized_eval = bool(vectorized_eval)
        self._dataloader: Optional[DataLoader] = None
        self._dataloader_iter = None
        self._fmodule = None
        self._net_obj = None
        self._subbatch_size = None if subbatch_size is None else int(subbatch_size)
        self._num_subbatches = None if num_subbatches is None else int(num_subbatches)

    # -- data plumbing -------------------------------------------------------

    @property
    def dataset(self) -> Dataset:
        return self._dataset

    def make_dataloader(self) -> DataLoader:
        g = torch.Generator()
        if self._seed is not None:
            g.manual_seed(int(self._seed))
        return DataLoader(self._dataset, batch_size=self._minibatch_size or 32, shuffle=True, generator=g)

    def get_minibatch(self):
        if self._dataloader is None:
            self._dataloader = self.make_dataloader()
            self._dataloader_iter = iter(self._dataloader)
        try:
            batch = next(self._dataloader_iter)
        except StopIteration:
            self._dataloader_iter = iter(self._dataloader)
            batch = next(self._dataloader_iter)
        return self._to_network_device(batch)

    def _to_network_device(self, batch):
        device = self.network_device
        if isinstance(batch, (list, tuple)):
            return type(batch)(x.to(device) if isinstance(x, torch.Tensor) else x for x in batch)
        return batch.to(device) if isinstance(batch, torch.Tensor) else batch

    # -- loss ----------------------------------------------------------------

    def loss(self, y_hat: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        """Public loss entry point (reference supervisedne.py:286): uses the
        provided loss_func or the overridable `_loss`."""
        return self._loss(y_hat, y)

    def _loss(self, y_hat: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        if self._loss_func is not None:
            return self._loss_func(y_hat, y)
        raise NotImplementedError("Provide loss_func or override _loss")

    def total_loss_of_net(self, network: nn.Module, batch) -> torch.Tensor:
        x, y = batch
        with torch.no_grad():
            return self._loss(network(x), y)

    # -- evaluation ----------------------------------------------------------

    def _evaluate_network(self, network: nn.Module) -> torch.Tensor:
        total = 0.0
        for _ in range(self._num_minibatches):
            batch = self.get_minibatch()
            total = total + self.total_loss_of_net(network, batch)
        return total / self._num_minibatches

    def _evaluate_batch(self, batch: SolutionBatch):
        if not (self._common_minibatch and self._vectorized_eval):
            super()._evaluate_batch(batch)
            return
        # common minibatch + whole-population vmapped forward. With
        # subbatch_size (reference supervisedne.py:334-348, the MNIST30K
        # config uses 50), each SUBBATCH of solutions shares one fresh
        # minibatch — a middle ground between per-solution minibatches
        # (high variance between solutions) and one global minibatch.
        if self._fmodule is None:
            self._net_obj = self._instantiate_net().to(self.network_device)
            self._fmodule = make_functional_module(self._net_obj)
        params = batch.access_values(keep_evals=True).to(self.network_device, torch.float32)
        n = len(batch)
        sub = self._subbatch_size
        if sub is None and self._num_subbatches is not None:
            sub = max(1, (n + self._num_subbatches - 1) // self._num_subbatches)
        losses = torch.zeros(n, dtype=torch.float32, device=self.network_device)
        # population_forward protocol: a net may provide its own
        # population-batched forward `population_forward(params (G, P),
        # x) -> (G, N, *out)` — e.g. convnets mapping member weights onto
        # channel GROUPS so the shared minibatch stays in one NCHW layout
        # end-to-end, skipping vmap's per-op member-dim reshapes (1.16x
        # on the MNIST30K benchmark, which is otherwise conv-FLOP-bound —
        # scripts/bench_supervised.py).
        pop_fwd = getattr(self._net_obj, "population_forward", None)
        with torch.no_grad():
            for _ in range(self._num_minibatches):
                for start in range(0, n, sub or n):
                    stop = min(start + (sub or n), n)
                    x, y = self.get_minibatch()

                    if pop_fwd is not None:
                        preds = pop_fwd(params[start:stop], x)
                        chunk = torch.func.vmap(lambda p: self._loss(p, y))(preds)
                    else:
                        def member_loss(flat):
                            return self._loss(self._fmodule._single(flat, x), y)

                        chunk = torch.func.vmap(member_loss, randomness="different")(params[start:stop])
                    losses[start:stop] = losses[start:stop] + chunk
        batch.set_evals((losses / self._num_minibatches).to(batch.device))
