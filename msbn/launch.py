"""msbn.launch — byte-compatible shim for the reference launch command
(/root/reference/README.md:98-100):

    python -m msbn.launch --nproc_per_node=N distributed_train.py --arg1 ...

Like the stock ``torch.distributed.launch`` (SURVEY.md §2.2 "launch.py"), it
passes ``--local-rank=<r>`` on each worker's argv (unless ``--use-env``) AND
sets the full env contract (LOCAL_RANK, RANK, WORLD_SIZE, MASTER_ADDR/PORT).
The worker script's argparse contract is README.md:15-19:

    parser.add_argument('--local_rank', type=int, default=0)

NOTE: argparse does NOT treat ``--local-rank`` and ``--local_rank`` as the
same option (verified on Python 3.10), so this shim injects the legacy
underscore spelling ``--local_rank=<r>`` that the reference README's
add_argument registers.  Workers written for modern torchrun (which injects
``--local-rank``) should register both spellings or read the ``LOCAL_RANK``
environment variable, which is always set.
"""

import sys

from msbn.run import parse_args, run


def main(argv=None) -> int:
    # use_env defaults to FALSE here: the legacy launcher injects a
    # local-rank argv (stock launch.py:145-149 behavior) — in the underscore
    # spelling the README's worker registers.
    args = parse_args(argv, use_env_default=False)
    args.legacy_underscore_flag = True
    return run(args)


if __name__ == "__main__":
    sys.exit(main())
