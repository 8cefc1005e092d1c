"""Abstract-dataflow feature-name DSL.

Parity target: reference sastvd/helpers/datasets.py:560-585 (parse_limits).
Feature names encode which abstract-dataflow subkeys are embedded and the
vocabulary limits, e.g.

  _ABS_DATAFLOW_datatype_all_limitall_1000_limitsubkeys_1000
  _ABS_DATAFLOW_api_datatype_literal_operator_all_limitall_5000_limitsubkeys_5000

Grammar: _ABS_DATAFLOW_<subkey>+_[all_]limitall_<N>_limitsubkeys_<M>
The model's input vocabulary is limitall + 2 (index 0 = "not a definition",
index 1 = UNKNOWN — dbize_absdf.py:35-42), which is what the datamodule
exposes as input_dim (datamodule.py:89-96).
"""

from __future__ import annotations

import re
from dataclasses import dataclass
from typing import List

ALL_SUBKEYS = ["api", "datatype", "literal", "operator"]


@dataclass
class FeatureSpec:
    name: str
    subkeys: List[str]
    all_subkeys: bool
    limit_all: int
    limit_subkeys: int

    @property
    def input_dim(self) -> int:
        return self.limit_all + 2


def parse_limits(feat: str) -> FeatureSpec:
    m = re.match(
        r"^_ABS_DATAFLOW(?P<subkeys>(?:_(?:api|datatype|literal|operator))+)"
        r"(?P<all>_all)?_limitall_(?P<limitall>\d+)_limitsubkeys_(?P<limitsubkeys>\d+)$",
        feat,
    )
    if not m:
        raise ValueError(f"unparseable feature name: {feat!r}")
    subkeys = [s for s in m.group("subkeys").split("_") if s]
    return FeatureSpec(
        name=feat,
        subkeys=subkeys,
        all_subkeys=m.group("all") is not None,
        limit_all=int(m.group("limitall")),
        limit_subkeys=int(m.group("limitsubkeys")),
    )
