import torch
import torch.nn as nn

from elasticdl_amd.common.constants import DistributionStrategy
from elasticdl_amd.layers.embedding import EdlEmbedding, bind_local_engine
from elasticdl_amd.ps.engine import PSEngine
from elasticdl_amd.utils.callbacks import LearningRateScheduler
from elasticdl_amd.utils.model_handler import ModelHandler


class Net(nn.Module):
    def __init__(self):
        super().__init__()
        self.big = nn.Embedding(100000, 64)    # 25.6 MB -> replaced
        self.small = nn.Embedding(10, 4)       # tiny -> kept
        self.fc = nn.Linear(64, 1)

    def forward(self, ids):
        return self.fc(self.big(ids).mean(1)).squeeze(-1)


def test_ps_handler_replaces_large_embeddings():
    h = ModelHandler.get_model_handler(DistributionStrategy.PARAMETER_SERVER)
    model = h.get_model_to_train(Net())
    assert isinstance(model.big, EdlEmbedding)
    assert isinstance(model.small, nn.Embedding)
    assert model.big.dim == 64
    assert model.big.name == "big"


def test_ps_handler_export_inverse():
    h = ModelHandler.get_model_handler(DistributionStrategy.PARAMETER_SERVER)
    model = h.get_model_to_train(Net())
    engine = PSEngine(device="cpu")
    engine.push_model({}, [])
    bind_local_engine(model, engine)
    ids = torch.tensor([[1, 2], [3, 4]])
    out = model(ids)  # creates rows lazily
    exported = h.get_model_to_export(model, engine)
    assert isinstance(exported.big, nn.Embedding)
    rows = engine.pull_embedding_vectors("big", torch.tensor([1, 2, 3, 4]),
                                         create=False)
    assert torch.allclose(exported.big.weight[1], rows[0])


def test_default_handler_noop():
    h = ModelHandler.get_model_handler(DistributionStrategy.ALLREDUCE)
    m = Net()
    assert h.get_model_to_train(m) is m


def test_lr_scheduler_torch_optimizer():
    model = nn.Linear(2, 1)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    sched = LearningRateScheduler(opt, lambda v: 0.5 if v >= 10 else 1.0)
    sched.on_train_batch_begin(0)
    assert opt.param_groups[0]["lr"] == 0.1
    sched.on_train_batch_begin(20)
    assert abs(opt.param_groups[0]["lr"] - 0.05) < 1e-9
    sched.on_train_batch_begin(0)
    assert opt.param_groups[0]["lr"] == 0.1


def test_lr_scheduler_distributed_optimizer():
    from elasticdl_amd.collective.distributed_optimizer import DistributedOptimizer

    model = nn.Linear(2, 1)
    opt = DistributedOptimizer(model, lr=0.2)
    sched = LearningRateScheduler(opt, lambda v: 0.1)
    sched.on_train_batch_begin(5)
    assert abs(opt.lr - 0.02) < 1e-9
