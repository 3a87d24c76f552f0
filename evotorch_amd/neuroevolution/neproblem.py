"""NEProblem: neuroevolution problems whose solutions are flat parameter
vectors of a neural network.

Reference parity: /root/reference/src/evotorch/neuroevolution/
neproblem.py:33-429 — `parameterize_net` loads a solution vector into a
cached network; fitness is `_evaluate_network` (override) or a provided
evaluation function; the network lives on the problem's aux device (the
GPU) while the population may live elsewhere.
"""

from typing import Callable, Optional, Union

import torch
from torch import nn

from ..core import Problem, Solution
from ..models import Policy, count_parameters, fill_parameters, str_to_net
from ..utils.misc import pass_info_if_needed

__all__ = ["NEProblem", "BaseNEProblem"]


class NEProblem(Problem):
    def __init__(
        self,
        objective_sense,
        network: Union[str, nn.Module, Callable[[], nn.Module]],
        network_eval_func: Optional[Callable] = None,
        *,
        network_args: Optional[dict] = None,
        initial_bounds=(-0.00001, 0.00001),
        eval_dtype=None,
        eval_data_length: int = 0,
        seed: Optional[int] = None,
        device=None,
        num_gpus_per_actor=None,  # accepted for API parity; SPMD topology comes from the launcher
        num_actors=None,
        actor_config=None,
        num_subbatches=None,
        subbatch_size=None,
        store_solution_stats: Optional[bool] = None,
    ):
        self._network_def = network
        self._network_args = dict(network_args or {})
        self._network_eval_func = network_eval_func
        self._instantiated_net: Optional[nn.Module] = None
        # instantiate once (on CPU) to learn the parameter count
        net = self._instantiate_net()
        solution_length = count_parameters(net)
        super().__init__(
            objective_sense,
            solution_length=solution_length,
            initial_bounds=initial_bounds,
            dtype=torch.float32,
            eval_dtype=eval_dtype,
            device=device,
            eval_data_length=eval_data_length,
            seed=seed,
            store_solution_stats=store_solution_stats,
            num_actors=num_actors,
            actor_config=actor_config,
            num_gpus_per_actor=num_gpus_per_actor,
            num_subbatches=num_subbatches,
            subbatch_size=subbatch_size,
        )
        self._instantiated_net = net.to(self.network_device)

    # -- network management ---------------------------------------------------

    @property
    def network_device(self) -> torch.device:
        """Where evaluated networks live: the aux (accelerator) device."""
        return self.aux_device

    def _network_constants(self) -> dict:
        """Named constants available to the string DSL; subclasses add
        e.g. obs_length / act_length."""
        return {}

    def _str_to_net(self, s: str) -> nn.Module:
        return str_to_net(s, **{**self._network_constants(), **self._network_args})

    def _instantiate_net(self) -> nn.Module:
        net = self._network_def
        if isinstance(net, str):
            module = self._str_to_net(net)
        elif isinstance(net, nn.Module):
            module = net
        elif callable(net):
            module = pass_info_if_needed(net, self._network_constants())(**self._network_args) if self._network_args or getattr(net, "__evotorch_pass_info__", False) else net()
        else:
            raise TypeError(f"Cannot instantiate a network from {type(net)}")
        return module

    @property
    def parameterized_net(self) -> Optional[nn.Module]:
        return self._instantiated_net

    @property
    def network_constants(self) -> dict:
        """Constants fed to str_to_net (obs_length etc.) — reference
        neproblem.py: network_constants."""
        return dict(self._network_constants())

    def parameterize_net(self, parameters: torch.Tensor) -> nn.Module:
        """Load a flat parameter vector into the cached network
        (reference neproblem.py:342)."""
        if self._instantiated_net is None:
            self._instantiated_net = self._instantiate_net().to(self.network_device)
        fill_parameters(self._instantiated_net, parameters.to(self.network_device))
        return self._instantiated_net

    def make_net(self, solution: Union[Solution, torch.Tensor]) -> nn.Module:
        """A fresh network carrying the solution's parameters."""
        import copy

        if isinstance(solution, Solution):
            values = torch.Tensor.as_subclass(solution.values, torch.Tensor)
        else:
            values = torch.as_tensor(solution)
        net = copy.deepcopy(self._instantiate_net())
        fill_parameters(net, values.to("cpu"))
        return net

    def make_functional_policy(self) -> Policy:
        """A Policy over this problem's architecture for whole-population
        batched forwards."""
        return Policy(self._instantiate_net())

    # -- evaluation -----------------------------------------------------------

    def _evaluate_network(self, network: nn.Module):
        """Subclass hook: fitness of one parameterized network."""
        raise NotImplementedError(
            "Override _evaluate_network, or provide network_eval_func"
        )

    def _evaluate(self, solution: Solution):
        net = self.parameterize_net(torch.Tensor.as_subclass(solution.values, torch.Tensor))
        fn = self._network_eval_func
        with torch.no_grad():  # fitness evaluation never needs autograd
            result = self._evaluate_network(net) if fn is None else fn(net)
        self._write_solution_result(solution, result)

    # -- pickling: do not ship the live network -------------------------------

    def _get_cloned_state(self, *, memo: dict) -> dict:
        net = self._instantiated_net
        self._instantiated_net = None
        try:
            state = super()._get_cloned_state(memo=memo)
        finally:
            self._instantiated_net = net
        return state


BaseNEProblem = NEProblem
