"""Trainer binary: gin-parse then train_eval_model().

Reference `bin/run_t2r_trainer.py:28-36`: the binary exposes only
--gin_configs / --gin_bindings; everything else is gin-bound.
"""

import argparse

from tensor2robot_amd import gin
from tensor2robot_amd.train import train_eval


def main(argv=None):
  parser = argparse.ArgumentParser(description=__doc__)
  parser.add_argument("--gin_configs", action="append", default=[],
                      help="Path(s) to gin config files.")
  parser.add_argument("--gin_bindings", action="append", default=[],
                      help="Individual gin bindings.")
  args = parser.parse_args(argv)
  gin.parse_config_files_and_bindings(args.gin_configs,
                                      "\n".join(args.gin_bindings))
  return train_eval.train_eval_model()


if __name__ == "__main__":
  main()
