"""Framework-wide default configuration constants.

Mirrors the reference's ``config.py`` (/root/reference/config.py:42-66): job
names, training defaults, evaluation/checkpoint/summary cadences and the
service-thread poll delay, so the `runner.py` CLI surface keeps the same
defaults as the reference.
"""

# ---------------------------------------------------------------------------- #
# Cluster / process-group defaults

default_ps_job_name = "ps"
default_wk_job_name = "workers"
default_ev_job_name = "eval"

# torch.distributed rendezvous defaults (MI355X-native single-node layout:
# one process per GPU over RCCL; gloo when no GPU is present)
default_master_addr = "127.0.0.1"
default_master_port = 29517

# ---------------------------------------------------------------------------- #
# Training defaults (reference config.py:47-51)

default_max_step          = 10000
default_learning_rate     = 1e-3
default_end_learning_rate = 1e-4
default_decay_step        = 10000
default_decay_rate        = 0.96

# ---------------------------------------------------------------------------- #
# Evaluation / checkpoint / summary defaults (reference config.py:54-61)

default_evaluation_file_name = "eval"
default_evaluation_delta     = -1
default_evaluation_period    = 10.
default_checkpoint_base_name = "model"
default_checkpoint_delta     = -1
default_checkpoint_period    = 120.
default_summary_delta        = -1
default_summary_period       = 30.

# ---------------------------------------------------------------------------- #
# Static configuration

thread_idle_delay = 1.  # Poll delay (s) of the eval/checkpoint/summary threads

# UDP-style lossy-gradient injection: chunk granularity in bytes, matching the
# reference's 65000-byte UDP payload chunks
# (reference tf_patches/patches/mpi_rendezvous_mgr.patch:563-592).
lossy_chunk_bytes = 65000
