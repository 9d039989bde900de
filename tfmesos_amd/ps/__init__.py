"""Parameter-server data plane.

* ``store``   — PS-side parameter store + fused optimizer apply
* ``replica`` — between-graph replica training (dense push/pull over
  RCCL collectives with the PS rank as root; sync + async modes)
* ``sparse``  — sparse embedding push/pull (gather/scatter-add kernels)
"""
