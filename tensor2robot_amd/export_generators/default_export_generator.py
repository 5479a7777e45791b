"""Default export generator: numpy + tf_example serving receivers.

Reference `export_generators/default_export_generator.py:33-133`. The
two receiver families are distinct code paths, as in the reference:

  * numpy receiver (:42-82): clients feed `{flat_key: ndarray}` for the
    required in-specs; the receiver validates/coerces against the spec
    and the preprocessing runs inside the exported servable (unless
    export_raw_receivers).
  * tf_example receiver (:84-133): clients feed serialized tf.Example
    bytes (one string feed per dataset_key); the spec-derived auto
    parser maps them to the numpy feed.

MI355X-native design note: a torchscript servable cannot execute a
protobuf parser, so the parse stage of the tf_example receiver runs
HOST-side from the parse contract EMBEDDED IN THE ARTIFACT — the
`assets.extra/t2r_assets.pbtxt` feature spec carries every name, dtype,
shape and image data_format the parser needs, and
`ExportedSavedModelPredictor.predict_serialized` reconstructs the exact
receiver from it with zero model code.  This mirrors the reference,
where the parse ran in the CPU portion of the serving graph while the
network ran on-device.  tf_example exports additionally ship their
warmup requests (`assets.extra/warmup_requests.tfrecord`) so a server
can replay representative traffic before going live (reference
:109-142).
"""

from __future__ import annotations

import os
from typing import Callable, Dict, List

import numpy as np
import torch

from tensor2robot_amd import gin
from tensor2robot_amd.export_generators import abstract_export_generator
from tensor2robot_amd.specs import tensorspec_utils as tsu

_tsu = tsu


@gin.configurable
class DefaultExportGenerator(abstract_export_generator.AbstractExportGenerator):
  """Numpy + tf_example receiver export (the standard servable)."""

  def create_serving_input_receiver_numpy_fn(self) -> Callable:
    """Numpy receiver: {flat_key: ndarray} -> validated tensor feed.

    Reference :42-82 — placeholders from the REQUIRED flat specs with
    dtype/shape coercion.
    """
    input_spec = self.serving_input_spec()

    def receiver(feed: Dict[str, np.ndarray]) -> Dict[str, torch.Tensor]:
      out: Dict[str, torch.Tensor] = {}
      for key, spec in input_spec.items():
        name = spec.name or key
        if key in feed:
          value = feed[key]
        elif name in feed:
          value = feed[name]
        else:
          raise ValueError(f"Missing required serving input {key!r}")
        t = torch.as_tensor(np.asarray(value))
        if spec.dtype.is_floating_point and not t.dtype.is_floating_point:
          t = t.float()
        out[key] = t
      return out

    return receiver

  def create_serving_input_receiver_tf_example_fn(self) -> Callable:
    """tf_example receiver: serialized Example bytes -> tensor feed.

    Reference :84-133 — one string feed per dataset_key, the
    spec-derived auto parser in front of the preprocessor.
    """
    from tensor2robot_amd.data import parser as parser_mod
    input_spec = self.serving_input_spec()
    parse_fn = parser_mod.create_parse_example_fn(input_spec)

    def receiver(serialized) -> Dict[str, torch.Tensor]:
      if not isinstance(serialized, dict):
        serialized = {"": list(serialized)}
      features, _ = parse_fn(serialized)
      return {k: v for k, v in features.items()
              if isinstance(v, torch.Tensor)}

    return receiver

  def export(self, model, export_root: str, global_step: int = 0,
             receiver_mode: str = "numpy",
             warmup_batch_sizes: List[int] = (1,)) -> str:
    """Export; tf_example mode embeds warmup requests in assets.extra."""
    final_dir = super().export(model, export_root,
                               global_step=global_step,
                               receiver_mode=receiver_mode)
    if receiver_mode == "tf_example":
      assets_dir = os.path.join(final_dir, tsu.EXTRA_ASSETS_DIRECTORY)
      self.create_warmup_requests_numpy(list(warmup_batch_sizes),
                                        assets_dir)
    return final_dir
