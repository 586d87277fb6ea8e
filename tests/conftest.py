import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that require an AMD GPU (MI355X); run with -m gpu")


def pytest_collection_modifyitems(config, items):
    """Auto-skip gpu-marked tests when no GPU is present, so a bare
    ``pytest tests/`` passes on CPU-only machines."""
    try:
        import torch
        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip = pytest.mark.skip(reason="no GPU on this machine")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def tiny_model_files(tmp_path):
    """Two-op chain model written as a pipedream .txt profile."""
    from ddls_amd.workloads import write_pipedream_txt
    nodes = [
        ("1", {"type": "Op1", "forward": 0.02, "backward": 0.04,
               "activation": 1e8, "parameter": 5e7}),
        ("2", {"type": "Op2", "forward": 0.03, "backward": 0.06,
               "activation": 2e8, "parameter": 1e8}),
    ]
    edges = [("1", "2")]
    d = tmp_path / "jobs"
    d.mkdir()
    write_pipedream_txt(str(d / "tiny.txt"), nodes, edges)
    return str(d)


def make_env(jobs_dir, replication=3, frac_dist=None, num_training_steps=10,
             max_partitions_per_op=16, min_quantum=0.01, interarrival=1000,
             workers=32, shape=(4, 4, 2)):
    from ddls_amd.envs import RampJobPartitioningEnvironment
    c, r, s = shape
    frac = frac_dist or {"_target_": "ddls_amd.distributions.Fixed", "val": 1.0}
    return RampJobPartitioningEnvironment(
        topology_config={"type": "ramp", "kwargs": {
            "num_communication_groups": c,
            "num_racks_per_communication_group": r,
            "num_servers_per_rack": s,
            "num_channels": 1,
            "total_node_bandwidth": 1.6e12,
            "intra_gpu_propagation_latency": 50e-9,
            "worker_io_latency": 100e-9}},
        node_config={"type_1": {"num_nodes": workers, "workers_config": [
            {"num_workers": 1, "worker": "ddls_amd.devices.A100"}]}},
        jobs_config={"path_to_files": jobs_dir,
                     "replication_factor": replication,
                     "job_sampling_mode": "remove",
                     "job_interarrival_time_dist": {
                         "_target_": "ddls_amd.distributions.Fixed",
                         "val": interarrival},
                     "max_acceptable_job_completion_time_frac_dist": frac,
                     "num_training_steps": num_training_steps},
        max_partitions_per_op=max_partitions_per_op,
        min_op_run_time_quantum=min_quantum,
        pad_obs_kwargs={"max_nodes": 150},
        max_simulation_run_time=1e6)
