"""tosem2021_amd — an MI355X-native framework with the capabilities of
openjamoses/TOSEM-2021-Replication.

The reference is the replication package of a TOSEM 2021 empirical study of
testing practices in nine ML systems (see /root/reference/README.md:1-15 and
SURVEY.md).  Its capability is a test-practice mining & analysis pipeline:

  corpus/    — L0/L1: project registry, corpus walker, test-file detection,
               repo-selection funnel          (ref: selection/Reposition/*)
  extract/   — L2: lift test cases + assertions out of Python/C++/TS suites
               (ref: the labeled rows of RQs/taxonomy_test2.csv)
  classify/  — L3: taxonomy labeling (19 strategies, 21 properties, workflow
               stages, test methods) — rule engine + learned classifier
  analyze/   — L4: RQ1/RQ3/RQ4 pivot tables + figures
               (ref: RQs/RQ1/Results/*, RQs/RQ3/*, RQs/RQ4/*)

The MI355X compute lane (the learned test classifier, trained on the study's
9,685 labeled rows) lives in:

  models/    — transformer test-case classifier (flagship: mltc-base)
  ops/       — hand-written gfx950 HIP kernels for the fused hot ops
  parallel/  — bucketed-allreduce data parallelism over RCCL/xGMI,
               fault-tolerant mining worker pool
  data/      — taxonomy datasets + synthetic benchmark data
  train.py   — trainer with checkpoint/resume, metrics, tracing
"""

__version__ = "0.1.0"
