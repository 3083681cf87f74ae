"""Model-zoo loading.

Mirrors common/model_utils.py:27-242: a model definition is a Python
module (file path or dotted module name) exposing the zoo contract —
``custom_model()``, ``loss()``, ``optimizer()``, ``eval_metrics_fn()``,
``feed()`` and optionally ``custom_data_reader()`` / ``callbacks()`` /
``synthetic_batch()``. Built-in zoo modules resolve by short name
(mnist, resnet50, wide_deep, deepfm, ...).
"""

import importlib
import importlib.util
import os
from dataclasses import dataclass, field
from typing import Callable, Optional

_BUILTIN = {
    "mnist": "elasticdl_amd.models.mnist",
    "resnet50": "elasticdl_amd.models.resnet",
    "wide_deep": "elasticdl_amd.models.wide_deep",
    "deepfm": "elasticdl_amd.models.deepfm",
    "dcn": "elasticdl_amd.models.dcn",
    "cifar10": "elasticdl_amd.models.cifar10",
    "iris": "elasticdl_amd.models.iris",
    "census_wide_deep": "elasticdl_amd.models.census_wide_deep",
    "census_dnn": "elasticdl_amd.models.census_dnn",
    "mobilenetv2": "elasticdl_amd.models.mobilenetv2",
    "heart": "elasticdl_amd.models.heart",
}


def _load_file(path: str):
    spec = importlib.util.spec_from_file_location(
        os.path.splitext(os.path.basename(path))[0], path
    )
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod


def load_module(model_def: str, model_zoo: str = ""):
    """Resolve a model definition to a module.

    Order (reference: common/model_utils.py:27-60, where model_def is a
    path inside the --model_zoo directory): built-in short name; a file
    inside ``model_zoo`` (``<zoo>/<model_def>`` or ``<zoo>/<model_def>.py``,
    dots treated as path separators so ``pkg.model`` finds
    ``<zoo>/pkg/model.py``); a standalone file path; a dotted module on
    sys.path (with ``model_zoo`` prepended when given).
    """
    if model_def in _BUILTIN:
        return importlib.import_module(_BUILTIN[model_def])
    if model_zoo:
        rel = model_def if model_def.endswith(".py") \
            else model_def.replace(".", os.sep) + ".py"
        for cand in (os.path.join(model_zoo, model_def),
                     os.path.join(model_zoo, rel)):
            if os.path.isfile(cand):
                return _load_file(cand)
        import sys
        if model_zoo not in sys.path:
            sys.path.insert(0, model_zoo)
    if os.path.isfile(model_def):
        return _load_file(model_def)
    return importlib.import_module(model_def)


@dataclass
class ModelSpec:
    module: object
    model_fn: Callable
    loss_fn: Callable
    optimizer_fn: Callable
    feed_fn: Optional[Callable] = None
    eval_metrics_fn: Optional[Callable] = None
    data_reader_fn: Optional[Callable] = None
    callbacks_fn: Optional[Callable] = None
    params: dict = field(default_factory=dict)

    def build_model(self):
        return self.model_fn(**self.params)


def get_model_spec(model_def: str, model_params: Optional[dict] = None,
                   model_zoo: str = "",
                   function_names: Optional[dict] = None) -> ModelSpec:
    """``function_names`` maps spec slots to custom function names in the
    module (reference flags --loss/--optimizer/--feed/... default to the
    conventional names), e.g. {"loss": "my_loss"}."""
    mod = load_module(model_def, model_zoo)
    names = dict(function_names or {})

    def pick(slot, default, required):
        name = names.get(slot) or default
        fn = getattr(mod, name, None)
        if fn is None and required:
            raise AttributeError(
                f"model zoo module {model_def!r} must define {name}()"
            )
        return fn

    return ModelSpec(
        module=mod,
        model_fn=pick("model", "custom_model", True),
        loss_fn=pick("loss", "loss", True),
        optimizer_fn=pick("optimizer", "optimizer", True),
        feed_fn=pick("feed", "feed", False),
        eval_metrics_fn=pick("eval_metrics_fn", "eval_metrics_fn", False),
        data_reader_fn=pick("custom_data_reader", "custom_data_reader",
                            False),
        callbacks_fn=pick("callbacks", "callbacks", False),
        params=model_params or {},
    )


def get_optimizer_info(opt) -> tuple:
    """Normalize a zoo optimizer() return into (opt_type, opt_args) for the
    PS CLI (reference: get_optimizer_info, common/model_utils.py:227)."""
    if isinstance(opt, tuple):
        return opt
    import torch

    if isinstance(opt, torch.optim.SGD):
        g = opt.param_groups[0]
        return (
            "momentum" if g.get("momentum", 0) else "sgd",
            f"learning_rate={g['lr']};momentum={g.get('momentum', 0)}"
            f";nesterov={str(bool(g.get('nesterov'))).lower()}",
        )
    if isinstance(opt, torch.optim.Adam):
        g = opt.param_groups[0]
        b1, b2 = g["betas"]
        return (
            "adam",
            f"learning_rate={g['lr']};beta_1={b1};beta_2={b2}"
            f";epsilon={g['eps']}",
        )
    if isinstance(opt, torch.optim.Adagrad):
        g = opt.param_groups[0]
        return ("adagrad", f"learning_rate={g['lr']};epsilon={g['eps']}")
    if isinstance(opt, torch.optim.RMSprop):
        g = opt.param_groups[0]
        return (
            "rmsprop",
            f"learning_rate={g['lr']};rho={g['alpha']}"
            f";momentum={g.get('momentum', 0)};epsilon={g['eps']}"
            f";centered={str(bool(g.get('centered'))).lower()}",
        )
    if isinstance(opt, torch.optim.Adadelta):
        g = opt.param_groups[0]
        return (
            "adadelta",
            f"learning_rate={g['lr']};rho={g['rho']};epsilon={g['eps']}",
        )
    if isinstance(opt, torch.optim.NAdam):
        g = opt.param_groups[0]
        b1, b2 = g["betas"]
        return (
            "nadam",
            f"learning_rate={g['lr']};beta_1={b1};beta_2={b2}"
            f";epsilon={g['eps']}",
        )
    if isinstance(opt, torch.optim.Adamax):
        g = opt.param_groups[0]
        b1, b2 = g["betas"]
        return (
            "adamax",
            f"learning_rate={g['lr']};beta_1={b1};beta_2={b2}"
            f";epsilon={g['eps']}",
        )
    raise ValueError(f"cannot map optimizer {type(opt)} to PS opt_type")
