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


class Config:
    _model_conf_keys = (
        "loss",
        "labels",
        "eval",
        "outputs_transform_for_loss",
        "outputs_transform_for_results",
    )

    models = {
        # ------------------------------------------------------ PhaseNet
        "phasenet": {
            "loss": partial(CELoss, weight=[[1], [1], [1]]),
            "inputs": [["z", "n", "e"]],
            "labels": [["non", "ppk", "spk"]],
            "eval": ["ppk", "spk"],
            "targets_transform_for_loss": None,
            "outputs_transform_for_loss": None,
            "outputs_transform_for_results": None,
        },
        # ------------------------------------------------------ EQTransformer
        "eqtransformer": {
            "loss": partial(BCELoss, weight=[[0.5], [1], [1]]),
            "inputs": [["z", "n", "e"]],
            "labels": [["det", "ppk", "spk"]],
            "eval": ["det", "ppk", "spk"],
            "targets_transform_for_loss": None,
            "outputs_transform_for_loss": None,
            "outputs_transform_for_results": None,
        },
        # ------------------------------------------------------ MagNet
        "magnet": {
            "loss": MousaviLoss,
            "inputs": [["z", "n", "e"]],
            "labels": ["emg"],
            "eval": ["emg"],
            "targets_transform_for_loss": None,
            "outputs_transform_for_loss": None,
            "outputs_transform_for_results": lambda x: x[:, 0].reshape(-1, 1),
        },
        # ------------------------------------------------------ BAZ Network
        "baz_network": {
            "loss": partial(CombinationLoss, losses=[MSELoss, MSELoss]),
            "inputs": [["z", "n", "e"]],
            "labels": ["baz"],
            "eval": ["baz"],
            "targets_transform_for_loss": lambda x: (
                (x * math.pi / 180).cos(),
                (x * math.pi / 180).sin(),
            ),
            "outputs_transform_for_loss": None,
            "outputs_transform_for_results": lambda x: torch.atan2(x[1], x[0])
            * 180
            / math.pi,
        },
        # ------------------------------------------------------ DiTingMotion
        "ditingmotion": {
            "loss": partial(CombinationLoss, losses=[FocalLoss, FocalLoss]),
            "inputs": [["z", "dz"]],
            "labels": ["clr", "pmp"],
            "eval": ["pmp"],
            "targets_transform_for_loss": None,
            "outputs_transform_for_loss": None,
            "outputs_transform_for_results": lambda xs: [x.softmax(-1) for x in xs],
        },
        # ------------------------------------------------------ SeisT dpk
        "seist_.*?_dpk.*": {
            "loss": partial(BCELoss, weight=[[0.5], [1], [1]]),
            "inputs": [["z", "n", "e"]],
            "labels": [["det", "ppk", "spk"]],
            "eval": ["det", "ppk", "spk"],
            "targets_transform_for_loss": None,
            "outputs_transform_for_loss": None,
            "outputs_transform_for_results": None,
        },
        # ------------------------------------------------------ SeisT pmp
        "seist_.*?_pmp": {
            "loss": partial(CELoss, weight=[1, 1]),
            "inputs": [["z", "n", "e"]],
            "labels": ["pmp"],
            "eval": ["pmp"],
            "targets_transform_for_loss": None,
            "outputs_transform_for_loss": None,
            "outputs_transform_for_results": None,
        },
        # ------------------------------------------------------ SeisT emg
        "seist_.*?_emg": {
            "loss": HuberLoss,
            "inputs": [["z", "n", "e"]],
            "labels": ["emg"],
            "eval": ["emg"],
            "targets_transform_for_loss": None,
            "outputs_transform_for_loss": None,
            "outputs_transform_for_results": None,
        },
        # ------------------------------------------------------ SeisT baz
        "seist_.*?_baz": {
            "loss": HuberLoss,
            "inputs": [["z", "n", "e"]],
            "labels": ["baz"],
            "eval": ["baz"],
            "targets_transform_for_loss": None,
            "outputs_transform_for_loss": None,
            "outputs_transform_for_results": None,
        },
        # ------------------------------------------------------ SeisT dis
        "seist_.*?_dis": {
            "loss": HuberLoss,
            "inputs": [["z", "n", "e"]],
            "labels": ["dis"],
            "eval": ["dis"],
            "targets_transform_for_loss": None,
            "outputs_transform_for_loss": None,
            "outputs_transform_for_results": None,
        },
    }

    _avl_metrics = ("precision", "recall", "f1", "mean", "rmse", "mae",
                    "mape", "r2")

    _avl_io_item_types = ("soft", "value", "onehot")

    _avl_io_items = {
        "z": {"type": "soft", "metrics": ["mean", "rmse", "mae"]},
        "n": {"type": "soft", "metrics": ["mean", "rmse", "mae"]},
        "e": {"type": "soft", "metrics": ["mean", "rmse", "mae"]},
        "dz": {"type": "soft", "metrics": ["mean", "rmse", "mae"]},
        "dn": {"type": "soft", "metrics": ["mean", "rmse", "mae"]},
        "de": {"type": "soft", "metrics": ["mean", "rmse", "mae"]},
        "non": {"type": "soft", "metrics": []},
        "det": {"type": "soft", "metrics": ["precision", "recall", "f1"]},
        "ppk": {"type": "soft",
                "metrics": ["precision", "recall", "f1", "mean", "rmse",
                            "mae", "mape"]},
        "spk": {"type": "soft",
                "metrics": ["precision", "recall", "f1", "mean", "rmse",
                            "mae", "mape"]},
        "ppk+": {"type": "soft", "metrics": []},
        "spk+": {"type": "soft", "metrics": []},
        "det+": {"type": "soft", "metrics": []},
        "ppks": {"type": "value",
                 "metrics": ["mean", "rmse", "mae", "mape", "r2"]},
        "spks": {"type": "value",
                 "metrics": ["mean", "rmse", "mae", "mape", "r2"]},
        "emg": {"type": "value", "metrics": ["mean", "rmse", "mae", "r2"]},
        "smg": {"type": "value", "metrics": ["mean", "rmse", "mae", "r2"]},
        "baz": {"type": "value", "metrics": ["mean", "rmse", "mae", "r2"]},
        "dis": {"type": "value", "metrics": ["mean", "rmse", "mae", "r2"]},
        "pmp": {"type": "onehot", "metrics": ["precision", "recall", "f1"],
                "num_classes": 2},
        "clr": {"type": "onehot", "metrics": ["precision", "recall", "f1"],
                "num_classes": 2},
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
