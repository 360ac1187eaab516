"""PS-side optimizers on a single flat fp32 parameter buffer.

The reference registers 10 TF-1.x optimizers (reference
sparkflow/tensorflow_async.py:19-30) and applies each incoming gradient
individually via ``optimizer.apply_gradients`` on the parameter-server
process (HogwildSparkModel.py:186-242).  Here the canonical weights live in
ONE contiguous fp32 tensor in HBM, so an optimizer step is a single
elementwise pass; on the MI355X the hot ones (adam, gradient_descent,
momentum, rmsprop) run as **one fused HIP kernel launch** over the flat
buffer (sparkflow_amd/ops/hip/optim.hip); the remaining names use the same
math in torch ops behind the identical registry names.

Update formulas follow TF 1.x semantics (epsilon placement, bias
correction, accumulator init) so converged behavior matches the reference.
"""
from __future__ import annotations

import math
from typing import Dict, Optional

import torch


def _tf_opts(options: Optional[Dict], name: str, learning_rate: float) -> Dict:
    # Reference default-options behavior: tensorflow_async.py:32-38.
    if options is None:
        options = {"learning_rate": learning_rate, "use_locking": False}
        if name == "momentum":
            options["momentum"] = 0.9
    options = dict(options)
    options.pop("use_locking", None)
    return options


class FlatOptimizer:
    """Base: ``apply(param, grad)`` performs one in-place update step."""

    name = "base"

    def __init__(self, numel: int, device, lr: float):
        self.numel = numel
        self.device = torch.device(device)
        self.lr = float(lr)
        self.t = 0  # step count (racy under Hogwild by design)

    def _zeros(self):
        return torch.zeros(self.numel, dtype=torch.float32, device=self.device)

    def _full(self, v):
        return torch.full((self.numel,), float(v), dtype=torch.float32,
                          device=self.device)

    def apply(self, param: torch.Tensor, grad: torch.Tensor) -> None:
        raise NotImplementedError

    def state_tensors(self):
        """Optimizer state for checkpoint/resume."""
        return {}

    def step_count(self) -> int:
        return self.t

    def set_step_count(self, t: int) -> None:
        self.t = int(t)

    def native_spec(self):
        """(opt_id, state tensors, lr, h1, h2, h3) for the C++ GIL-free
        PS session loop (ops/hip/dist_session.hip), or None when this
        optimizer has no fused native apply."""
        return None


class GradientDescent(FlatOptimizer):
    name = "gradient_descent"

    def __init__(self, numel, device, learning_rate=0.001, **_):
        super().__init__(numel, device, learning_rate)

    def apply(self, param, grad):
        from sparkflow_amd import ops

        if ops.use_hip(param):
            ops.hip.fused_sgd(param, grad, self.lr)
        else:
            param.add_(grad, alpha=-self.lr)

    def native_spec(self):
        return (0, [], self.lr, 0.0, 0.0, 0.0)


class Momentum(FlatOptimizer):
    name = "momentum"

    def __init__(self, numel, device, learning_rate=0.001, momentum=0.9,
                 use_nesterov=False, **_):
        super().__init__(numel, device, learning_rate)
        self.momentum = float(momentum)
        self.nesterov = bool(use_nesterov)
        self.acc = self._zeros()

    def apply(self, param, grad):
        from sparkflow_amd import ops

        if ops.use_hip(param):
            ops.hip.fused_momentum(param, grad, self.acc, self.lr,
                                   self.momentum, self.nesterov)
            return
        self.acc.mul_(self.momentum).add_(grad)
        if self.nesterov:
            param.add_(grad, alpha=-self.lr).add_(self.acc,
                                                  alpha=-self.lr * self.momentum)
        else:
            param.add_(self.acc, alpha=-self.lr)

    def state_tensors(self):
        return {"acc": self.acc}

    def native_spec(self):
        return (2, [self.acc], self.lr, self.momentum,
                1.0 if self.nesterov else 0.0, 0.0)


class Adam(FlatOptimizer):
    name = "adam"

    def __init__(self, numel, device, learning_rate=0.001, beta1=0.9,
                 beta2=0.999, epsilon=1e-8, **_):
        super().__init__(numel, device, learning_rate)
        self.beta1, self.beta2, self.eps = float(beta1), float(beta2), float(epsilon)
        self.m = self._zeros()
        self.v = self._zeros()
        # GPU path: [t, lr_t for step t+1, block ticket, pad] on-device so
        # an apply is ONE hipGraph-capturable launch (the apply kernel's
        # last block advances t and precomputes the next lr_t).
        self.dev_state = None
        if self.device.type == "cuda":
            self.dev_state = torch.zeros(4, dtype=torch.float32,
                                         device=self.device)
            self._seed_dev_lr_t()

    def _seed_dev_lr_t(self):
        t_next = float(self.t) + 1.0
        lr_t = self.lr * math.sqrt(1.0 - self.beta2 ** t_next) / (
            1.0 - self.beta1 ** t_next)
        self.dev_state[0] = float(self.t)
        self.dev_state[1] = lr_t
        self.dev_state[2] = 0.0

    def apply(self, param, grad):
        from sparkflow_amd import ops

        if ops.use_hip(param):
            ops.hip.fused_adam_dev(param, grad, self.m, self.v,
                                   self.dev_state, self.lr, self.beta1,
                                   self.beta2, self.eps)
            return
        self.t += 1
        # TF1 Adam: lr_t = lr * sqrt(1-b2^t) / (1-b1^t); p -= lr_t * m/(sqrt(v)+eps)
        lr_t = self.lr * math.sqrt(1.0 - self.beta2 ** self.t) / (
            1.0 - self.beta1 ** self.t)
        self.m.mul_(self.beta1).add_(grad, alpha=1.0 - self.beta1)
        self.v.mul_(self.beta2).addcmul_(grad, grad, value=1.0 - self.beta2)
        param.addcdiv_(self.m, self.v.sqrt().add_(self.eps), value=-lr_t)

    def state_tensors(self):
        return {"m": self.m, "v": self.v}

    def apply_from_workspaces(self, param, ws_list, offsets):
        """Fused split-K-reduce + Adam apply: `ws_list` are the raw wgrad
        workspaces (deferred reduces, ops.functional.defer_wgrad) whose
        planes tile the flat buffer at `offsets`.  One launch replaces
        the per-layer reduces, pack_segments, and fused_adam_dev; math
        and add order are identical (optim.hip reduce_apply_adam)."""
        from sparkflow_amd import ops

        ops.hip.reduce_apply_adam(ws_list, offsets, param, self.m, self.v,
                                  self.dev_state, self.lr, self.beta1,
                                  self.beta2, self.eps)

    def native_spec(self):
        if self.dev_state is None:
            return None
        return (1, [self.m, self.v, self.dev_state], self.lr, self.beta1,
                self.beta2, self.eps)

    def step_count(self) -> int:
        if self.dev_state is not None:
            return int(self.dev_state[0].item())
        return self.t

    def set_step_count(self, t: int) -> None:
        self.t = int(t)
        if self.dev_state is not None:
            self._seed_dev_lr_t()


class RMSProp(FlatOptimizer):
    name = "rmsprop"

    def __init__(self, numel, device, learning_rate=0.001, decay=0.9,
                 momentum=0.0, epsilon=1e-10, centered=False, **_):
        super().__init__(numel, device, learning_rate)
        self.decay, self.momentum, self.eps = float(decay), float(momentum), float(epsilon)
        self.centered = bool(centered)
        self.ms = self._zeros()
        self.mom = self._zeros()
        self.mg = self._zeros() if self.centered else None

    def apply(self, param, grad):
        from sparkflow_amd import ops

        if ops.use_hip(param) and not self.centered:
            ops.hip.fused_rmsprop(param, grad, self.ms, self.mom, self.lr,
                                  self.decay, self.momentum, self.eps)
            return
        self.ms.mul_(self.decay).addcmul_(grad, grad, value=1.0 - self.decay)
        if self.centered:
            self.mg.mul_(self.decay).add_(grad, alpha=1.0 - self.decay)
            denom = (self.ms - self.mg * self.mg).add_(self.eps).sqrt_()
        else:
            # TF rmsprop: epsilon inside the sqrt
            denom = (self.ms + self.eps).sqrt_()
        self.mom.mul_(self.momentum).addcdiv_(grad, denom, value=self.lr)
        param.sub_(self.mom)

    def state_tensors(self):
        s = {"ms": self.ms, "mom": self.mom}
        if self.mg is not None:
            s["mg"] = self.mg
        return s

    def native_spec(self):
        if self.centered:
            return None
        return (3, [self.ms, self.mom], self.lr, self.decay, self.momentum,
                self.eps)


class Adadelta(FlatOptimizer):
    name = "adadelta"

    def __init__(self, numel, device, learning_rate=0.001, rho=0.95,
                 epsilon=1e-8, **_):
        super().__init__(numel, device, learning_rate)
        self.rho, self.eps = float(rho), float(epsilon)
        self.accum = self._zeros()
        self.accum_update = self._zeros()

    def apply(self, param, grad):
        self.accum.mul_(self.rho).addcmul_(grad, grad, value=1.0 - self.rho)
        update = (self.accum_update + self.eps).sqrt_() \
            .div_((self.accum + self.eps).sqrt_()).mul_(grad)
        self.accum_update.mul_(self.rho).addcmul_(update, update,
                                                  value=1.0 - self.rho)
        param.add_(update, alpha=-self.lr)

    def state_tensors(self):
        return {"accum": self.accum, "accum_update": self.accum_update}


class Adagrad(FlatOptimizer):
    name = "adagrad"

    def __init__(self, numel, device, learning_rate=0.001,
                 initial_accumulator=0.1, initial_accumulator_value=None, **_):
        super().__init__(numel, device, learning_rate)
        init = initial_accumulator_value if initial_accumulator_value is not None \
            else initial_accumulator
        self.accum = self._full(init)

    def apply(self, param, grad):
        self.accum.addcmul_(grad, grad, value=1.0)
        param.addcdiv_(grad, self.accum.sqrt(), value=-self.lr)

    def state_tensors(self):
        return {"accum": self.accum}


class AdagradDA(FlatOptimizer):
    name = "adagrad_da"

    def __init__(self, numel, device, learning_rate=0.001,
                 initial_gradient_squared_accumulator_value=0.1,
                 l1_regularization_strength=0.0,
                 l2_regularization_strength=0.0, **_):
        super().__init__(numel, device, learning_rate)
        self.g_acc = self._zeros()
        self.gg_acc = self._full(initial_gradient_squared_accumulator_value)
        self.l1 = float(l1_regularization_strength)
        self.l2 = float(l2_regularization_strength)

    def apply(self, param, grad):
        self.t += 1
        self.g_acc.add_(grad)
        self.gg_acc.addcmul_(grad, grad, value=1.0)
        t = float(self.t)
        denom = self.gg_acc.sqrt().add_(self.l2 * t * self.lr)
        if self.l1 > 0:
            num = torch.clamp(self.g_acc.abs() - self.l1 * t, min=0.0)
            param.copy_(-self.lr * torch.sign(self.g_acc) * num / denom)
        else:
            param.copy_(-self.lr * self.g_acc / denom)

    def state_tensors(self):
        return {"g_acc": self.g_acc, "gg_acc": self.gg_acc}


class Ftrl(FlatOptimizer):
    name = "ftrl"

    def __init__(self, numel, device, learning_rate=0.001,
                 learning_rate_power=-0.5, initial_accumulator_value=0.1,
                 l1_regularization_strength=0.0,
                 l2_regularization_strength=0.0, **_):
        super().__init__(numel, device, learning_rate)
        self.lr_power = float(learning_rate_power)
        self.n = self._full(initial_accumulator_value)
        self.z = self._zeros()
        self.l1 = float(l1_regularization_strength)
        self.l2 = float(l2_regularization_strength)

    def apply(self, param, grad):
        n_new = self.n + grad * grad
        sigma = (n_new.pow(-self.lr_power) - self.n.pow(-self.lr_power)) / self.lr
        self.z.add_(grad).sub_(sigma * param)
        self.n.copy_(n_new)
        denom = n_new.pow(-self.lr_power) / self.lr + 2.0 * self.l2
        if self.l1 > 0:
            num = torch.sign(self.z) * self.l1 - self.z
            param.copy_(torch.where(self.z.abs() > self.l1, num / denom,
                                    torch.zeros_like(param)))
        else:
            param.copy_(-self.z / denom)

    def state_tensors(self):
        return {"n": self.n, "z": self.z}


def _prox(param, lr, l1, l2):
    if l1 > 0:
        param.copy_(torch.sign(param) *
                    torch.clamp(param.abs() - lr * l1, min=0.0))
    if l2 > 0:
        param.div_(1.0 + lr * l2)


class ProximalGradientDescent(FlatOptimizer):
    name = "proximal_gradient_descent"

    def __init__(self, numel, device, learning_rate=0.001,
                 l1_regularization_strength=0.0,
                 l2_regularization_strength=0.0, **_):
        super().__init__(numel, device, learning_rate)
        self.l1 = float(l1_regularization_strength)
        self.l2 = float(l2_regularization_strength)

    def apply(self, param, grad):
        param.add_(grad, alpha=-self.lr)
        _prox(param, self.lr, self.l1, self.l2)


class ProximalAdagrad(FlatOptimizer):
    name = "proximal_adagrad"

    def __init__(self, numel, device, learning_rate=0.001,
                 initial_accumulator_value=0.1,
                 l1_regularization_strength=0.0,
                 l2_regularization_strength=0.0, **_):
        super().__init__(numel, device, learning_rate)
        self.accum = self._full(initial_accumulator_value)
        self.l1 = float(l1_regularization_strength)
        self.l2 = float(l2_regularization_strength)

    def apply(self, param, grad):
        self.accum.addcmul_(grad, grad, value=1.0)
        lr_t = self.lr / self.accum.sqrt()
        param.sub_(lr_t * grad)
        if self.l1 > 0:
            param.copy_(torch.sign(param) *
                        torch.clamp(param.abs() - lr_t * self.l1, min=0.0))
        if self.l2 > 0:
            param.div_(1.0 + lr_t * self.l2)

    def state_tensors(self):
        return {"accum": self.accum}


REGISTRY = {
    cls.name: cls
    for cls in [Adam, RMSProp, Momentum, Adadelta, Adagrad, GradientDescent,
                AdagradDA, Ftrl, ProximalAdagrad, ProximalGradientDescent]
}


def make_optimizer(name: str, numel: int, device,
                   learning_rate: float = 0.001,
                   options: Optional[Dict] = None) -> FlatOptimizer:
    """Build a flat-buffer optimizer by reference registry name.

    Unknown names fall back to gradient_descent, matching the reference
    (tensorflow_async.py:40-42).
    """
    opts = _tf_opts(options, name, learning_rate)
    cls = REGISTRY.get(name, GradientDescent)
    return cls(numel, device, **opts)
