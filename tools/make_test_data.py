"""Regenerates checked-in test fixtures (test_data/)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from tensor2robot_amd.research.pose_env import pose_env_models

root = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "test_data")
os.makedirs(root, exist_ok=True)
path = pose_env_models.generate_test_tfrecord(
    os.path.join(root, "pose_env_test_data.tfrecord"), num_records=24, seed=7)
print("wrote", path)
