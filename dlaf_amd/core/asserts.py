"""Assertion levels (reference ``common/assert.h:1-163``).

Three levels, enabled by ``DLAF_ASSERT_LEVEL`` (default 1):
  0 - nothing checked;
  1 - DLAF_ASSERT: cheap precondition checks (always recommended);
  2 - DLAF_ASSERT_MODERATE: O(cheap) consistency checks;
  3 - DLAF_ASSERT_HEAVY: expensive validation (test builds).

Unlike Python ``assert`` these are NOT elided under ``python -O``, and a
failure raises a single consistent exception type so distributed callers
can translate it into a rank-consistent abort.
"""

from __future__ import annotations

import os


class DlafAssertError(AssertionError):
    pass


def _level() -> int:
    try:
        return int(os.environ.get("DLAF_ASSERT_LEVEL", "1"))
    except ValueError:
        return 1


def dlaf_assert(cond: bool, *msg) -> None:
    if _level() >= 1 and not cond:
        raise DlafAssertError(" ".join(str(m) for m in msg) or "DLAF_ASSERT")


def dlaf_assert_moderate(cond: bool, *msg) -> None:
    if _level() >= 2 and not cond:
        raise DlafAssertError(
            " ".join(str(m) for m in msg) or "DLAF_ASSERT_MODERATE")


def dlaf_assert_heavy(cond_fn, *msg) -> None:
    """``cond_fn`` is a callable so the (expensive) check only evaluates at
    level >= 3."""
    if _level() >= 3 and not cond_fn():
        raise DlafAssertError(
            " ".join(str(m) for m in msg) or "DLAF_ASSERT_HEAVY")
