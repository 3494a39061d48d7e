# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Opt-in cross-rank op-sequence consistency checking.

The reference coordinator validates on every negotiation cycle that all
ranks submitted matching ops — same name order, shape, dtype, neighbor
sets — and its stall report names the missing ranks (reference:
bluefog/common/operations.cc:293-433, 853-1115). This framework's default
program-order contract skips that coordination entirely: a rank
submitting ops in a different order (multi-model scripts, data-dependent
control flow) deadlocks inside RCCL with no diagnosis.

``BLUEFOG_CHECK_CONSISTENCY=K`` (K ops per block; ``1``/``true`` → 16)
turns on the debug checker: every rank appends a descriptor per submitted
op (name, payload bytes, shape/dtype/src/dst when provided) and, after
each block of K, publishes the block to the TCP store and compares all
ranks' digests for that block. A mismatch raises immediately and NAMES the
diverging rank(s) and the first differing op, instead of hanging for the
watchdog to notice.
"""

import hashlib
import json
import os
import threading
from typing import List, Optional

from bluefog_amd.utils.logging import get_logger

_VERIFY_TIMEOUT_S = float(os.environ.get("BLUEFOG_CHECK_CONSISTENCY_TIMEOUT", "60"))


def _block_size() -> int:
    raw = os.environ.get("BLUEFOG_CHECK_CONSISTENCY", "").strip().lower()
    if raw in ("", "0", "false", "off"):
        return 0
    if raw in ("1", "true", "on"):
        return 16
    try:
        return max(1, int(raw))
    except ValueError:
        return 16


class ConsistencyChecker:
    def __init__(self):
        self._k = _block_size()
        self._lock = threading.Lock()
        self._descs: List[str] = []
        self._block = 0

    @property
    def enabled(self) -> bool:
        return self._k > 0

    def record(self, name: str, nbytes: Optional[int], fingerprint: str = "",
               detail: str = "") -> None:
        """Append one op descriptor; verify when the block is full. Called
        on the submitting thread — verification is synchronous, as the
        reference's negotiation was.

        ``fingerprint`` must be RANK-INVARIANT (shape/dtype/root): it goes
        into the cross-rank digest. ``detail`` may be rank-relative
        (src/dst sets, payload bytes that scale with the neighbor count) —
        it is shown in diagnostics but never compared."""
        if not self.enabled:
            return
        # the auto-name ordinal IS the order; identical programs produce
        # identical ordinals. nbytes stays out of the digest (it scales
        # with the per-rank neighbor count).
        desc = f"{name}|{fingerprint}||{nbytes}|{detail}"
        with self._lock:
            self._descs.append(desc)
            if len(self._descs) < self._k:
                return
            descs, block = self._descs, self._block
            self._descs, self._block = [], self._block + 1
        self._verify(block, descs)

    def flush(self) -> None:
        """Verify any partial block (e.g. at shutdown or on demand)."""
        if not self.enabled:
            return
        with self._lock:
            if not self._descs:
                return
            descs, block = self._descs, self._block
            self._descs, self._block = [], self._block + 1
        self._verify(block, descs)

    def _verify(self, block: int, descs: List[str]) -> None:
        from bluefog_amd.ops.context import ctx

        c = ctx()
        if c.size() == 1:
            return
        me = c.rank()
        payload = json.dumps(descs).encode()
        invariant = [d.split("||", 1)[0] for d in descs]
        digest = hashlib.sha1(json.dumps(invariant).encode()).hexdigest()
        c.store.set(f"cc/{block}/{me}", digest.encode() + b"\n" + payload)
        keys = [f"cc/{block}/{r}" for r in range(c.size())]
        try:
            c.store.wait(keys, timeout_s=_VERIFY_TIMEOUT_S)
        except Exception:
            missing = [r for r in range(c.size())
                       if not c.store.check([f"cc/{block}/{r}"])]
            raise RuntimeError(
                f"bluefog_amd consistency check: rank(s) {missing} did not "
                f"submit op block {block} within {_VERIFY_TIMEOUT_S:.0f}s "
                f"(they are behind or hung). This rank's block starts with "
                f"{descs[0]!r}."
            ) from None
        blobs = c.store.multi_get(keys)
        digests = [b.split(b"\n", 1)[0].decode() for b in blobs]
        if len(set(digests)) == 1:
            return
        # majority digest defines the reference order; diverging ranks are
        # named with the first differing op
        counts = {}
        for d in digests:
            counts[d] = counts.get(d, 0) + 1
        # majority digest defines the reference order; ties go to the
        # lowest rank holding the digest (rank 0 is the coordinator in the
        # reference design, operations.cc:880-898)
        majority = max(counts, key=lambda d: (counts[d], -digests.index(d)))
        maj_descs = json.loads(
            next(b for b in blobs if b.startswith(majority.encode())).split(b"\n", 1)[1]
        )
        lines = []
        for r, (d, b) in enumerate(zip(digests, blobs)):
            if d == majority:
                continue
            r_descs = json.loads(b.split(b"\n", 1)[1])
            idx = next(
                (i for i, (a, x) in enumerate(zip(maj_descs, r_descs))
                 if a.split("||", 1)[0] != x.split("||", 1)[0]),
                min(len(maj_descs), len(r_descs)),
            )
            got = r_descs[idx] if idx < len(r_descs) else "<nothing>"
            exp = maj_descs[idx] if idx < len(maj_descs) else "<nothing>"
            lines.append(
                f"  rank {r}: op #{block * self._k + idx} is {got!r}, "
                f"majority submitted {exp!r}"
            )
        msg = (
            f"bluefog_amd consistency check FAILED at op block {block}: "
            f"rank(s) diverge from the majority op order "
            f"(reference analog: ConstructResponse validation, "
            f"operations.cc:293-384):\n" + "\n".join(lines)
        )
        get_logger().error(msg)
        raise RuntimeError(msg)


_checker: Optional[ConsistencyChecker] = None
_checker_lock = threading.Lock()


def checker() -> ConsistencyChecker:
    global _checker
    with _checker_lock:
        if _checker is None:
            _checker = ConsistencyChecker()
        return _checker


def _reset_for_tests() -> None:
    global _checker
    with _checker_lock:
        _checker = None
