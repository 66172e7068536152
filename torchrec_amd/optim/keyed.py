"""FQN-keyed optimizer surface.

Reference parity: torchrec/optim/keyed.py (KeyedOptimizer :34,
CombinedOptimizer :317, KeyedOptimizerWrapper :436) and torchrec/optim/fused.py
(FusedOptimizer :17 — step is a no-op; the update happened inside the TBE
backward kernel).
"""

from __future__ import annotations

from typing import Any, Callable, Dict, List, Mapping, Optional, Set, Tuple, Union

import torch
from torch import optim


class KeyedOptimizer(optim.Optimizer):
    """Optimizer whose state_dict is keyed by parameter FQN.

    params: FQN -> parameter/buffer. state/param_groups as in torch.optim.
    """

    def __init__(
        self,
        params: Mapping[str, Union[torch.Tensor, Any]],
        state: Mapping[Any, Any],
        param_groups: List[Dict[str, Any]],
    ) -> None:
        # NOTE: deliberately does NOT call super().__init__ (same as reference)
        torch._C._log_api_usage_once("torchrec_amd.optim.KeyedOptimizer")
        self.params = params
        self.state = state
        self.param_groups = param_groups
        self.defaults: Dict[str, Any] = {"_save_param_groups": False}

    def state_dict(self) -> Dict[str, Any]:
        param_to_key = {v: k for k, v in self.params.items()}
        state = {param_to_key[p]: v for p, v in self.state.items() if p in param_to_key}
        return {"state": state, "param_groups": []}

    def load_state_dict(self, state_dict: Mapping[str, Any]) -> None:
        new_state = state_dict["state"]
        for key, param in self.params.items():
            if key not in new_state:
                continue
            if param in self.state:
                cur = self.state[param]
                for sk, sv in new_state[key].items():
                    if sk in cur and isinstance(cur[sk], torch.Tensor):
                        cur[sk].detach().copy_(sv)
                    else:
                        cur[sk] = sv
            else:
                self.state[param] = new_state[key]

    def add_param_group(self, param_group: Any) -> None:
        raise NotImplementedError()

    def init_state(self, sparse_grad_parameter_names: Optional[Set[str]] = None) -> None:
        """Run a zero-grad step so state tensors materialize for checkpoint load."""
        for key, t in self.params.items():
            if isinstance(t, torch.Tensor) and t.requires_grad:
                t.grad = torch.zeros_like(t)
        self.step(closure=None)

    def save_param_groups(self, save: bool) -> None:
        self.defaults["_save_param_groups"] = save

    def step(self, closure: Any = None) -> None:
        raise NotImplementedError()

    def zero_grad(self, set_to_none: bool = False) -> None:
        for t in self.params.values():
            if isinstance(t, torch.Tensor) and t.grad is not None:
                if set_to_none:
                    t.grad = None
                else:
                    t.grad.zero_()


class FusedOptimizer(KeyedOptimizer):
    """step/zero_grad are no-ops — the update runs inside the backward kernel
    (reference optim/fused.py:17)."""

    def step(self, closure: Any = None) -> None:
        pass

    def zero_grad(self, set_to_none: bool = False) -> None:
        pass


class CombinedOptimizer(KeyedOptimizer):
    """Combines optimizers with key prefixes (reference keyed.py:317)."""

    def __init__(
        self, optims: List[Union[KeyedOptimizer, Tuple[str, KeyedOptimizer]]]
    ) -> None:
        self.defaults: Dict[str, Any] = {}
        self._optims: List[Tuple[str, KeyedOptimizer]] = []
        for e in optims:
            if isinstance(e, tuple):
                self._optims.append(e)
            else:
                self._optims.append(("", e))

    @property
    def optimizers(self) -> List[Tuple[str, KeyedOptimizer]]:
        return self._optims

    @staticmethod
    def prepend_opt_key(name: str, opt_key: str) -> str:
        if not opt_key:
            return name
        return f"{opt_key}.{name}"

    @property
    def param_groups(self) -> List[Dict[str, Any]]:
        return [pg for _, o in self._optims for pg in o.param_groups]

    @property
    def params(self) -> Mapping[str, Any]:
        ret = {}
        for key, o in self._optims:
            for n, p in o.params.items():
                ret[CombinedOptimizer.prepend_opt_key(n, key)] = p
        return ret

    @property
    def state(self) -> Mapping[torch.Tensor, Any]:
        ret = {}
        for _, o in self._optims:
            ret.update(o.state)
        return ret

    def state_dict(self) -> Dict[str, Any]:
        state = {}
        for key, o in self._optims:
            sd = o.state_dict()
            for n, v in sd["state"].items():
                state[CombinedOptimizer.prepend_opt_key(n, key)] = v
        return {"state": state, "param_groups": []}

    def load_state_dict(self, state_dict: Mapping[str, Any]) -> None:
        for key, o in self._optims:
            prefix = f"{key}." if key else ""
            sub = {
                n[len(prefix):]: v
                for n, v in state_dict["state"].items()
                if n.startswith(prefix)
            }
            o.load_state_dict({"state": sub, "param_groups": []})

    def step(self, closure: Any = None) -> None:
        for _, o in self._optims:
            o.step(closure=closure)

    def zero_grad(self, set_to_none: bool = False) -> None:
        for _, o in self._optims:
            o.zero_grad(set_to_none=set_to_none)

    def save_param_groups(self, save: bool) -> None:
        for _, o in self._optims:
            o.save_param_groups(save)


class KeyedOptimizerWrapper(KeyedOptimizer):
    """Wrap a torch optimizer factory over named params (reference keyed.py:436)."""

    def __init__(
        self,
        params: Mapping[str, torch.Tensor],
        optim_factory: Callable[[List[torch.Tensor]], optim.Optimizer],
    ) -> None:
        self._optimizer = optim_factory(list(params.values()))
        super().__init__(params, self._optimizer.state, self._optimizer.param_groups)

    def step(self, closure: Any = None) -> None:
        self._optimizer.step(closure=closure)

    def zero_grad(self, set_to_none: bool = False) -> None:
        self._optimizer.zero_grad(set_to_none=set_to_none)
