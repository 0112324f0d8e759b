"""msbn.launch — byte-compatible shim for the reference launch command
(/root/reference/README.md:98-100):

    python -m msbn.launch --nproc_per_node=N distributed_train.py --arg1 ...

Like the stock ``torch.distributed.launch`` (SURVEY.md §2.2 "launch.py"), it
passes ``--local-rank=<r>`` on each worker's argv (unless ``--use-env``) AND
sets the full env contract (LOCAL_RANK, RANK, WORLD_SIZE, MASTER_ADDR/PORT).
The worker script's argparse contract is README.md:15-19:

    parser.add_argument('--local_rank', type=int, default=0)

(argparse treats ``--local-rank`` and ``--local_rank`` as the same option).
"""

import sys

from msbn.run import parse_args, run


def main(argv=None) -> int:
    # use_env defaults to FALSE here: the legacy launcher injects
    # --local-rank=<r> argv (stock launch.py:145-149 behavior).
    args = parse_args(argv, use_env_default=False)
    return run(args)


if __name__ == "__main__":
    sys.exit(main())
