"""Graph/model build helpers — parity with reference sparkflow/graph_utils.py.

``build_graph(func)`` keeps the reference's contract (graph_utils.py:6-15):
run a user model-builder function and return a JSON string that fully
describes architecture + initial weights.  Here ``func`` returns a PyTorch
``nn.Module`` (see sparkflow_amd.utils.serialization for the module
convention) instead of building a TF graph.

The ``build_*_config`` helpers (graph_utils.py:18-47) are preserved with the
identical signatures and TF-style option names; the optimizer registry in
``sparkflow_amd.spark.async_dl`` / ``sparkflow_amd.parallel.optimizers``
understands these names and maps them onto the MI355X-native optimizers
(fused single-launch HIP Adam/SGD on the PS GPU).
"""
import json


def build_graph(func):
    """
    :param func: function returning an ``nn.Module`` (forward(x)->out,
        compute_loss(out, target)->scalar).  Weights initialized inside the
        call are captured so every process reconstructs the same init.
    :return: JSON string artifact (the ``tensorflowGraph`` Param payload).
    """
    from sparkflow_amd.utils.serialization import module_to_json

    module = func()
    return module_to_json(func, module)


def generate_config(**kwargs):
    return json.dumps(kwargs)


def build_adam_config(learning_rate=0.001, beta1=0.9, beta2=0.999, epsilon=1e-8,
                      use_locking=False):
    return generate_config(learning_rate=learning_rate, beta1=beta1, beta2=beta2,
                           epsilon=epsilon, use_locking=use_locking)


def build_rmsprop_config(learning_rate=0.001, decay=0.9, momentum=0.0,
                         epsilon=1e-10, use_locking=False, centered=False):
    return generate_config(learning_rate=learning_rate, decay=decay,
                           momentum=momentum, epsilon=epsilon,
                           use_locking=use_locking, centered=centered)


def build_momentum_config(learning_rate=0.001, momentum=0.9, use_locking=False,
                          use_nesterov=False):
    return generate_config(learning_rate=learning_rate, momentum=momentum,
                           use_locking=use_locking, use_nesterov=use_nesterov)


def build_adadelta_config(learning_rate=0.001, rho=0.95, epsilon=1e-8,
                          use_locking=False):
    return generate_config(learning_rate=learning_rate, rho=rho, epsilon=epsilon,
                           use_locking=use_locking)


def build_adagrad_config(learning_rate=0.001, initial_accumulator=0.1,
                         use_locking=False):
    return generate_config(learning_rate=learning_rate,
                           initial_accumulator=initial_accumulator,
                           use_locking=use_locking)


def build_gradient_descent(learning_rate=0.001, use_locking=False):
    return generate_config(learning_rate=learning_rate, use_locking=use_locking)
