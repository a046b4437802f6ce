"""Collective-primitive identifiers.

Mirrors the reference's primitive numbering (reference: commu.py:28-35,
csrc/include/trans.h:27-36) so strategy files and entry-point flags are
interchangeable, but here ALLGATHER / ALLTOALL / REDUCESCATTER are fully
implemented rather than declared-only.
"""

from __future__ import annotations

import enum


class Primitive(enum.IntEnum):
    ALLREDUCE = 0
    REDUCE = 1
    BROADCAST = 2
    ALLGATHER = 3
    ALLTOALL = 4
    REDUCESCATTER = 5
    DETECT = 6
    PROFILE = 7

    # Sentinel used as entry_point when a pre-synthesized strategy file is
    # supplied (reference: adapcc.py entry_point=-1).
    @classmethod
    def from_entry_point(cls, value: int) -> "Primitive | None":
        if value == -1:
            return None
        return cls(value)


# Backwards-compatible aliases matching the reference's (mis)spellings so
# user code written against AdapCC ports without edits.
ALLREDUCE = Primitive.ALLREDUCE
REDUCE = Primitive.REDUCE
BROADCAST = Primitive.BROADCAST
BOARDCAST = Primitive.BROADCAST  # reference spelling (commu.py:30)
ALLGATHER = Primitive.ALLGATHER
ALLTOALL = Primitive.ALLTOALL
REDUCESCATTER = Primitive.REDUCESCATTER
DETECT = Primitive.DETECT
PROFILE = Primitive.PROFILE
