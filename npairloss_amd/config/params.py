"""Typed configuration objects mirroring the reference's protobuf schemas.

- `NPairLossConfig` <-> `NPairLossParameter` (reference caffe.proto:2-23,
  extension field 8866720): margins, identsn/diffsn order-statistic
  selectors, and the (region x method) mining enums for positives (ap) and
  negatives (an).
- `SolverConfig` <-> the Caffe SolverParameter subset the reference's
  usage/solver.prototxt:1-17 uses (SGD + step LR policy + snapshotting).

Both parse from prototxt text via `config.prototxt`.
"""

from __future__ import annotations

import enum
from dataclasses import dataclass, field, asdict
from typing import List, Optional

import numpy as np

from .prototxt import Message, parse_prototxt


class MiningRegion(enum.IntEnum):
    """caffe.proto:7-10 — whether thresholds are per-query or batch-global."""

    GLOBAL = 0
    LOCAL = 1


class MiningMethod(enum.IntEnum):
    """caffe.proto:11-17 — threshold source + comparison direction."""

    HARD = 0
    EASY = 1
    RAND = 2  # select ALL pairs (reference .cu:88-89,109-110)
    RELATIVE_HARD = 3
    RELATIVE_EASY = 4


def _enum_from(value, enum_cls):
    if isinstance(value, enum_cls):
        return value
    if isinstance(value, str):
        return enum_cls[value]
    return enum_cls(int(value))


@dataclass
class NPairLossConfig:
    """Mirrors NPairLossParameter (reference caffe.proto:2-23), same defaults.

    identsn/diffsn semantics (reference .cu:282-305, 313-336): for the
    RELATIVE_* methods the threshold is an order statistic of the ascending-
    sorted positive (identsn) / negative (diffsn) similarity list:
      sn >= 0 : index = len-1 - int(sn)        (absolute count from the top)
      sn <  0 : index = int(len-1 + sn*len)    (fraction from the top)
    A selected threshold value < 0 is clamped to -inf, i.e. select-all
    (reference .cu:288,303,319,334).
    """

    margin_ident: float = 0.0
    margin_diff: float = 0.0
    identsn: float = -1.0
    diffsn: float = -1.0
    ap_mining_region: MiningRegion = MiningRegion.LOCAL
    ap_mining_method: MiningMethod = MiningMethod.RAND
    an_mining_region: MiningRegion = MiningRegion.LOCAL
    an_mining_method: MiningMethod = MiningMethod.RAND

    def __post_init__(self) -> None:
        self.ap_mining_region = _enum_from(self.ap_mining_region, MiningRegion)
        self.ap_mining_method = _enum_from(self.ap_mining_method, MiningMethod)
        self.an_mining_region = _enum_from(self.an_mining_region, MiningRegion)
        self.an_mining_method = _enum_from(self.an_mining_method, MiningMethod)
        # Store scalars at float32 precision like the proto (so identsn=-0.3
        # reproduces the reference's 0.30000001192... arithmetic), but keep
        # the sign of -0.0: -0.0 >= 0 must stay True (production config uses
        # identsn: -0.0 => top-of-list absolute index, def.prototxt:139).
        self.margin_ident = float(np.float32(self.margin_ident))
        self.margin_diff = float(np.float32(self.margin_diff))
        self.identsn = float(np.float32(self.identsn))
        self.diffsn = float(np.float32(self.diffsn))

    @classmethod
    def from_message(cls, msg: Message) -> "NPairLossConfig":
        kwargs = {}
        for name in (
            "margin_ident",
            "margin_diff",
            "identsn",
            "diffsn",
            "ap_mining_region",
            "ap_mining_method",
            "an_mining_region",
            "an_mining_method",
        ):
            if msg.has(name):
                kwargs[name] = msg.get(name)
        return cls(**kwargs)

    @classmethod
    def from_prototxt(cls, text: str) -> "NPairLossConfig":
        msg = parse_prototxt(text)
        inner = msg.get("npair_loss_param")
        return cls.from_message(inner if inner is not None else msg)

    def to_dict(self) -> dict:
        d = asdict(self)
        for k in ("ap_mining_region", "an_mining_region"):
            d[k] = MiningRegion(d[k]).name
        for k in ("ap_mining_method", "an_mining_method"):
            d[k] = MiningMethod(d[k]).name
        return d


@dataclass
class SolverConfig:
    """The SolverParameter subset of usage/solver.prototxt:1-17."""

    net: Optional[str] = None
    test_iter: int = 0
    test_interval: int = 0
    test_initialization: bool = True
    display: int = 0
    average_loss: int = 1
    base_lr: float = 0.01
    lr_policy: str = "fixed"
    stepsize: int = 0
    gamma: float = 1.0
    power: float = 1.0
    max_iter: int = 0
    iter_size: int = 1
    momentum: float = 0.0
    weight_decay: float = 0.0
    snapshot: int = 0
    snapshot_prefix: str = ""
    solver_mode: str = "GPU"
    random_seed: Optional[int] = None

    @classmethod
    def from_message(cls, msg: Message) -> "SolverConfig":
        kwargs = {}
        for name in cls.__dataclass_fields__:
            if msg.has(name):
                kwargs[name] = msg.get(name)
        return cls(**kwargs)

    @classmethod
    def from_prototxt(cls, text: str) -> "SolverConfig":
        return cls.from_message(parse_prototxt(text))

    def lr_at(self, it: int) -> float:
        """Learning rate at iteration `it` per Caffe's lr_policy semantics."""
        if self.lr_policy == "fixed":
            return self.base_lr
        if self.lr_policy == "step":
            return self.base_lr * (self.gamma ** (it // max(1, self.stepsize)))
        if self.lr_policy == "exp":
            return self.base_lr * (self.gamma ** it)
        if self.lr_policy == "inv":
            return self.base_lr * (1.0 + self.gamma * it) ** (-self.power)
        if self.lr_policy == "poly":
            return self.base_lr * (1.0 - it / max(1, self.max_iter)) ** self.power
        raise ValueError("unsupported lr_policy: %r" % self.lr_policy)


def parse_solver_prototxt(text: str) -> SolverConfig:
    return SolverConfig.from_prototxt(text)


@dataclass
class LayerSpec:
    """One `layer { ... }` block of a net prototxt (name/type/bottoms/tops +
    the raw Message for type-specific params)."""

    name: str
    type: str
    bottoms: List[str] = field(default_factory=list)
    tops: List[str] = field(default_factory=list)
    phase: Optional[str] = None  # "TRAIN" / "TEST" / None (both)
    loss_weights: List[float] = field(default_factory=list)
    raw: Optional[Message] = None

    @classmethod
    def from_message(cls, msg: Message) -> "LayerSpec":
        phase = None
        inc = msg.get("include")
        if inc is not None and inc.has("phase"):
            phase = str(inc.get("phase"))
        return cls(
            name=str(msg.get("name", "")),
            type=str(msg.get("type", "")),
            bottoms=[str(b) for b in msg.get_all("bottom")],
            tops=[str(t) for t in msg.get_all("top")],
            phase=phase,
            loss_weights=[float(w) for w in msg.get_all("loss_weight")],
            raw=msg,
        )


@dataclass
class NetConfig:
    name: str
    layers: List[LayerSpec]

    def layers_for_phase(self, phase: str) -> List[LayerSpec]:
        return [l for l in self.layers if l.phase is None or l.phase == phase]

    def find(self, type_name: str) -> List[LayerSpec]:
        return [l for l in self.layers if l.type == type_name]


def parse_net_prototxt(text: str) -> NetConfig:
    msg = parse_prototxt(text)
    layers = [LayerSpec.from_message(m) for m in msg.get_all("layer")]
    return NetConfig(name=str(msg.get("name", "")), layers=layers)
