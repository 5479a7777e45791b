"""Converts legacy pickle-based assets to t2r_assets.pbtxt assets.

Reference `utils/convert_pkl_assets_to_proto_assets.py:35-61`: reads the
exported servable's `assets.extra` directory, loads the legacy
`input_specifications.pkl` (feature/label specs) and optional global
step, and writes the pbtxt `t2r_assets.pbtxt` next to it.
"""

import argparse
import os

from tensor2robot_amd.specs import tensorspec_utils as tsu


def convert(assets_filepath: str) -> str:
  """Converts pickle assets under `assets_filepath`; returns pbtxt path."""
  pkl_path = os.path.join(assets_filepath, tsu.INPUT_SPEC_PKL_FILENAME)
  if not os.path.exists(pkl_path):
    # The reference also accepted `input_specs.pkl`.
    alt = os.path.join(assets_filepath, "input_specs.pkl")
    if os.path.exists(alt):
      pkl_path = alt
    else:
      raise ValueError(f"No file exists for {pkl_path}.")
  out_path = os.path.join(assets_filepath, tsu.T2R_ASSETS_FILENAME)
  tsu.convert_pkl_assets_to_proto_assets(pkl_path, out_path)
  return out_path


def main(argv=None):
  parser = argparse.ArgumentParser(description=__doc__)
  parser.add_argument("--assets_filepath", required=True,
                      help="Exported savedmodel assets directory.")
  args = parser.parse_args(argv)
  return convert(args.assets_filepath)


if __name__ == "__main__":
  main()
