"""Run-mode constants (the reference's tf.estimator.ModeKeys
analog; see abstract_model.py model_fn mode handling)."""

TRAIN = "train"
EVAL = "eval"
PREDICT = "predict"

ALL_MODES = (TRAIN, EVAL, PREDICT)


def validate(mode):
  if mode not in ALL_MODES:
    raise ValueError(f"Unknown mode {mode!r}; expected one of {ALL_MODES}")
  return mode
