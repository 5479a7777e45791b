"""Actor binary: gin-parse then collect_eval_loop().

Reference `bin/run_collect_eval.py:40-44`.
"""

import argparse

from tensor2robot_amd import gin
from tensor2robot_amd.utils import continuous_collect_eval


def main(argv=None):
  parser = argparse.ArgumentParser(description=__doc__)
  parser.add_argument("--gin_configs", action="append", default=[])
  parser.add_argument("--gin_bindings", action="append", default=[])
  args = parser.parse_args(argv)
  gin.parse_config_files_and_bindings(args.gin_configs,
                                      "\n".join(args.gin_bindings))
  return continuous_collect_eval.collect_eval_loop()


if __name__ == "__main__":
  main()
