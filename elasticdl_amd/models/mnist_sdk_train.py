"""SDK-style zoo module: a user-owned elastic training loop.

The reference's second zoo contract (model_zoo/mnist/mnist_train_tfv2.py:
21-40): instead of exposing custom_model()/loss()/..., the module exposes
``train(data_stream, elastic_controller)`` and drives its own loop; the
controller supplies elastic allreduce, batch retry and shard accounting.
"""

import torch
import torch.nn as nn

from elasticdl_amd.models.mnist import MnistCNN, synthetic_batch  # noqa: F401


def create_model_and_optimizer(master_client, batch_size):
    from elasticdl_amd.collective.controller import ElasticAllReduceController
    from elasticdl_amd.collective.distributed_optimizer import DistributedOptimizer

    device = "cuda" if torch.cuda.is_available() else "cpu"
    model = MnistCNN().to(device)
    if device == "cuda":
        model = model.to(torch.bfloat16)
    opt = DistributedOptimizer(model, lr=0.01, momentum=0.9)
    controller = ElasticAllReduceController(
        master_client, model, opt, batch_size=batch_size
    )
    return model, opt, controller, device


def train(data_stream, elastic_controller, model, optimizer, device="cpu"):
    """data_stream yields (images, labels) minibatches; the controller
    wraps each step with elastic retry + shard accounting."""

    @elastic_controller.elastic_run
    def train_one(x, y):
        optimizer.zero_grad()
        out = model(x.to(device))
        loss = nn.functional.cross_entropy(out.float(), y.to(device))
        loss.backward()
        optimizer.step()
        return loss

    losses = []
    for x, y in data_stream:
        losses.append(float(train_one(x, y)))
    return losses


def custom_data_reader(data_origin: str = ""):
    from elasticdl_amd.models.mnist import custom_data_reader as base

    return base(data_origin)
