"""Global except hook — fail the whole job on any rank's exception.

Reference: chainermn/global_except_hook.py (~L1-60): replace
``sys.excepthook`` to print a rank-prefixed traceback and MPI_Abort so one
rank's Python exception kills the job instead of deadlocking collectives.

No MPI here: after printing, the hook destroys the torch.distributed
process group (waking peers with an error where possible) and hard-exits.
Peers blocked in RCCL collectives are reaped by the launcher (torchrun
kills the job when any local rank dies nonzero).

Install with ``chainermn_amd.global_except_hook.add_hook()``; disable via
env CHAINERMN_AMD_DISABLE_GLOBAL_EXCEPT_HOOK=1.
"""

import os
import sys
import traceback

_hook_installed = False


def add_hook():
    global _hook_installed
    if _hook_installed:
        return
    if os.environ.get('CHAINERMN_AMD_DISABLE_GLOBAL_EXCEPT_HOOK') == '1':
        return
    _hook_installed = True
    prev = sys.excepthook

    def _hook(exc_type, exc_value, exc_tb):
        rank = os.environ.get('RANK', '?')
        lines = traceback.format_exception(exc_type, exc_value, exc_tb)
        msg = ''.join('[rank {}] {}'.format(rank, ln) for ln in lines)
        sys.stderr.write(msg)
        sys.stderr.flush()
        try:
            prev(exc_type, exc_value, exc_tb)
        except Exception:
            pass
        try:
            import torch.distributed as dist
            if dist.is_available() and dist.is_initialized():
                dist.destroy_process_group()
        except Exception:
            pass
        sys.stderr.write(
            '[rank {}] aborting job after uncaught exception\n'.format(rank))
        sys.stderr.flush()
        os._exit(1)

    sys.excepthook = _hook
