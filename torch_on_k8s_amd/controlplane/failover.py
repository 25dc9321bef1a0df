"""Failure classification and failover actions.

Reference: controllers/common/failover.go —
  * permanent exit codes 1,2,126,127,128,139 (:64-99)
  * transient (retryable) 130,137,143; 138 = SIGUSR1 user-defined retryable
  * retryable kill reasons OOMKilled/Killed/Evicted/... (:106-113)
  * actions: Recreate vs in-place restart (:43-48,117-264). The reference
    leans on OpenKruise CRR for in-place container restart; on a single
    node both collapse to "restart the process", with RECREATE releasing
    and re-acquiring GPU slots and IN_PLACE keeping the same slots
    (SURVEY.md §7 'hard parts': recreate-only is acceptable; we keep both
    so elastic restarts preserve slot locality).
"""
from __future__ import annotations

import enum

PERMANENT_EXIT_CODES = {1, 2, 126, 127, 128, 139}
TRANSIENT_EXIT_CODES = {129, 130, 137, 143}  # SIGHUP/SIGINT/SIGKILL/SIGTERM
USER_RETRYABLE_EXIT_CODES = {138}            # SIGUSR1 convention
RETRYABLE_REASONS = {"OOMKilled", "Killed", "Evicted",
                     "UnexpectedAdmissionError", "NodeLost"}


class FailoverAction(str, enum.Enum):
    NONE = "none"                  # permanent failure: surface it
    RECREATE = "recreate"
    IN_PLACE_RESTART = "in-place"


def exit_code_retryable(code: int | None, reason: str = "") -> bool:
    if reason in RETRYABLE_REASONS:
        return True
    if code is None:
        return False
    if code in TRANSIENT_EXIT_CODES or code in USER_RETRYABLE_EXIT_CODES:
        return True
    if code in PERMANENT_EXIT_CODES:
        return False
    # negative = killed by signal (python subprocess convention)
    if code < 0:
        return (128 - code) not in PERMANENT_EXIT_CODES
    # 128-255 range = signal exits; the env contract treats them retryable
    return 128 < code < 256
