"""Test-side alias: the vhost-user master lives in the package
(oim_amd.bench.vhost_client) so the benchmark harness shares it."""

from oim_amd.bench.vhost_client import *  # noqa: F401,F403
from oim_amd.bench.vhost_client import (  # noqa: F401
    QSIZE, QUEUE, DESC_NEXT, DESC_WRITE, GPA_BASE, REQ_OFF, RESP_OFF,
    DATA_IN_OFF, DATA_OUT_OFF, INDIRECT_OFF, UADDR_BASE,
    VhostUserMaster, ScsiResult, read_blk_config,
    GET_CONFIG, GET_QUEUE_NUM, GET_VRING_BASE, FEAT_VERSION_1,
    FEAT_INDIRECT, FEAT_PROTOCOL,
)
