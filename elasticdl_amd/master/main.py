"""Master entrypoint (reference: master/main.py:20-28)."""

import sys

from elasticdl_amd.common.args import parse_master_args
from elasticdl_amd.master.master import Master


def main(argv=None) -> int:
    args = parse_master_args(argv)
    master = Master(args)
    master.prepare()
    return master.run()


if __name__ == "__main__":
    sys.exit(main())
