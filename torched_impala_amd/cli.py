"""CLI driver: `python -m torched_impala_amd.cli
--env synthetic-atari --num-actors 32 ...` — train or test."""
from __future__ import annotations

import logging

from torched_impala_amd.config import flags_from_args
from torched_impala_amd import learner


def main(argv=None):
    logging.basicConfig(
        level=logging.INFO,
        format="%(asctime)s %(levelname).1s %(name)s: %(message)s")
    flags = flags_from_args(argv)
    if flags.mode == "train":
        stats = learner.train(flags)
        logging.info("done: %s", stats)
    else:
        learner.test(flags)


if __name__ == "__main__":
    main()
