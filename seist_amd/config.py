"""Declarative task/model configuration.

Semantic parity with /root/reference/config.py: a regex-keyed ``models``
dict binding each registered model name to its loss constructor, input
groups, label groups, eval tasks and optional transforms; an io-item
registry typing every named tensor (soft / value / onehot) with its metric
set; and import-time validation (`check_and_init`). This is the mechanism
that lets one train/val/test loop drive all 21 models across 5 tasks.
"""

import math
import re
from collections import defaultdict
from functools import partial
from typing import Any

import torch

from .models import (
    BCELoss,
    CELoss,
    CombinationLoss,
    FocalLoss,
    HuberLoss,
    MousaviLoss,
    MSELoss,
    get_model_list,
)

_ZNE = ["z", "n", "e"]


def _bind(loss, inputs, labels, eval_tasks, tgt_loss=None, out_loss=None,
          out_results=None):
    """One model->task binding (the reference spells these dicts out
    longhand; the keys and values are identical)."""
    return {
        "loss": loss,
        "inputs": inputs,
        "labels": labels,
        "eval": eval_tasks,
        "targets_transform_for_loss": tgt_loss,
        "outputs_transform_for_loss": out_loss,
        "outputs_transform_for_results": out_results,
    }


def _deg2trig(x):
    rad = x * math.pi / 180
    return rad.cos(), rad.sin()


def _io(kind, metrics, num_classes=None):
    item = {"type": kind, "metrics": list(metrics)}
    if num_classes is not None:
        item["num_classes"] = num_classes
    return item


# metric bundles shared by io items
_WAVE_M = ("mean", "rmse", "mae")
_PICK_M = ("precision", "recall", "f1", "mean", "rmse", "mae", "mape")
_DET_M = ("precision", "recall", "f1")
_IDX_M = ("mean", "rmse", "mae", "mape", "r2")
_REG_M = ("mean", "rmse", "mae", "r2")


class Config:
    _model_conf_keys = (
        "loss",
        "labels",
        "eval",
        "outputs_transform_for_loss",
        "outputs_transform_for_results",
    )

    models = {
        "phasenet": _bind(
            partial(CELoss, weight=[[1], [1], [1]]),
            inputs=[_ZNE], labels=[["non", "ppk", "spk"]],
            eval_tasks=["ppk", "spk"]),
        "eqtransformer": _bind(
            partial(BCELoss, weight=[[0.5], [1], [1]]),
            inputs=[_ZNE], labels=[["det", "ppk", "spk"]],
            eval_tasks=["det", "ppk", "spk"]),
        "magnet": _bind(
            MousaviLoss, inputs=[_ZNE], labels=["emg"], eval_tasks=["emg"],
            out_results=lambda x: x[:, 0].reshape(-1, 1)),
        "baz_network": _bind(
            partial(CombinationLoss, losses=[MSELoss, MSELoss]),
            inputs=[_ZNE], labels=["baz"], eval_tasks=["baz"],
            tgt_loss=_deg2trig,
            out_results=lambda x: torch.atan2(x[1], x[0]) * 180 / math.pi),
        "ditingmotion": _bind(
            partial(CombinationLoss, losses=[FocalLoss, FocalLoss]),
            inputs=[["z", "dz"]], labels=["clr", "pmp"], eval_tasks=["pmp"],
            out_results=lambda xs: [x.softmax(-1) for x in xs]),
        "seist_.*?_dpk.*": _bind(
            partial(BCELoss, weight=[[0.5], [1], [1]]),
            inputs=[_ZNE], labels=[["det", "ppk", "spk"]],
            eval_tasks=["det", "ppk", "spk"]),
        "seist_.*?_pmp": _bind(
            partial(CELoss, weight=[1, 1]),
            inputs=[_ZNE], labels=["pmp"], eval_tasks=["pmp"]),
        "seist_.*?_emg": _bind(
            HuberLoss, inputs=[_ZNE], labels=["emg"], eval_tasks=["emg"]),
        "seist_.*?_baz": _bind(
            HuberLoss, inputs=[_ZNE], labels=["baz"], eval_tasks=["baz"]),
        "seist_.*?_dis": _bind(
            HuberLoss, inputs=[_ZNE], labels=["dis"], eval_tasks=["dis"]),
    }

    _avl_metrics = ("precision", "recall", "f1", "mean", "rmse", "mae",
                    "mape", "r2")

    _avl_io_item_types = ("soft", "value", "onehot")

    _avl_io_items = {
        **{ch: _io("soft", _WAVE_M) for ch in ("z", "n", "e",
                                               "dz", "dn", "de")},
        "non": _io("soft", ()),
        "det": _io("soft", _DET_M),
        "ppk": _io("soft", _PICK_M),
        "spk": _io("soft", _PICK_M),
        "ppk+": _io("soft", ()),
        "spk+": _io("soft", ()),
        "det+": _io("soft", ()),
        "ppks": _io("value", _IDX_M),
        "spks": _io("value", _IDX_M),
        "emg": _io("value", _REG_M),
        "smg": _io("value", _REG_M),
        "baz": _io("value", _REG_M),
        "dis": _io("value", _REG_M),
        "pmp": _io("onehot", _DET_M, num_classes=2),
        "clr": _io("onehot", _DET_M, num_classes=2),
    }

    # ------------------------------------------------------------------

    @classmethod
    def check_and_init(cls):
        cls._type_to_ioitems = defaultdict(list)
        for k, v in cls._avl_io_items.items():
            cls._type_to_ioitems[v["type"]].append(k)

        unused = list(cls.models)
        for reg_name in get_model_list():
            for re_name in cls.models:
                if re.findall(re_name, reg_name) and re_name in unused:
                    unused.remove(re_name)
        if unused:
            print(f"Useless configurations: {unused}")

        for name, conf in cls.models.items():
            missing = set(cls._model_conf_keys) - set(conf)
            if missing:
                raise Exception(f"Model:'{name}'  Missing keys:{missing}")
            labels = sum([g if isinstance(g, (tuple, list)) else [g]
                          for g in conf["labels"]], [])
            unknown = set(labels) - set(cls._avl_io_items)
            if unknown:
                raise NotImplementedError(
                    f"Model:'{name}'  Unknown labels:{unknown}")
            inputs = sum([g if isinstance(g, (tuple, list)) else [g]
                          for g in conf["inputs"]], [])
            unknown = set(inputs) - set(cls._avl_io_items)
            if unknown:
                raise NotImplementedError(
                    f"Model:'{name}'  Unknown inputs:{unknown}")
            unknown = set(conf["eval"]) - set(cls._avl_io_items)
            if unknown:
                raise NotImplementedError(
                    f"Model:'{name}'  Unknown tasks:{unknown}")

        for k, v in cls._avl_io_items.items():
            if v["type"] not in cls._avl_io_item_types:
                raise NotImplementedError(
                    f"Unknown item type: {v['type']}, item: {k}")
            unknown = set(v["metrics"]) - set(cls._avl_metrics)
            if unknown:
                raise NotImplementedError(
                    f"Unknown metrics:{unknown} , item: {k}")

    @classmethod
    def get_io_items(cls, type: str = None) -> list:
        if type is None:
            return list(cls._avl_io_items)
        return cls._type_to_ioitems[type]

    @classmethod
    def get_type(cls, name: str) -> str:
        return cls._avl_io_items[name]["type"]

    @classmethod
    def get_num_classes(cls, name: str) -> int:
        if name not in cls._avl_io_items:
            raise ValueError(f"Name {name} not exists.")
        item_type = cls._avl_io_items[name]["type"]
        if item_type != "onehot":
            raise Exception(f"Type of item '{name}' is '{item_type}'.")
        return cls._avl_io_items[name]["num_classes"]

    @classmethod
    def get_model_config(cls, model_name: str) -> dict:
        registered = get_model_list()
        if model_name not in registered:
            raise NotImplementedError(
                f"Unknown model:'{model_name}', registered: {registered}")
        matches = [re_name for re_name in cls.models
                   if re.findall(re_name, model_name)]
        if len(matches) < 1:
            raise Exception(f"Missing configuration of model {model_name}")
        if len(matches) > 1:
            raise Exception(
                f"Model {model_name} matches multiple configuration items: "
                f"{matches}")
        return cls.models[matches[0]]

    @classmethod
    def get_model_config_(cls, model_name: str, *attrs) -> Any:
        conf = cls.get_model_config(model_name)
        vals = []
        for attr in attrs:
            if attr not in conf:
                raise Exception(
                    f"Unknown attribute:'{attr}', supported: {list(conf)}")
            vals.append(conf[attr])
        return vals[0] if len(vals) == 1 else tuple(vals)

    @classmethod
    def get_num_inchannels(cls, model_name: str) -> int:
        for inp in cls.get_model_config_(model_name, "inputs"):
            if isinstance(inp, (list, tuple)):
                if cls._avl_io_items[inp[0]]["type"] == "soft":
                    return len(inp)
        raise Exception(f"Incorrect input channels. Model:{model_name}")

    @classmethod
    def get_metrics(cls, item_name: str) -> list:
        if item_name not in cls._avl_io_items:
            raise Exception(
                f"Unknown item:'{item_name}', supported: "
                f"{list(cls._avl_io_items)}")
        return cls._avl_io_items[item_name]["metrics"]

    @classmethod
    def get_loss(cls, model_name: str):
        return cls.get_model_config(model_name)["loss"]()


Config.check_and_init()
