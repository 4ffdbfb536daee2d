"""oracle — CPU restatement of BigStitcher-Spark's hot-path arithmetic.

TEST INFRASTRUCTURE ONLY. Only tests/, __graft_entry__.smoke() and
bench.py's cpu_baseline leg may import or execute anything in this package,
and only as the checker / reported CPU baseline — never as the thing
measured or shipped. The product path (bigstitcher_spark_amd) must never
route through this package; it fails loudly when its HIP extension is
missing.

PARITY STATUS: **parity unpinned** against the true reference. The
reference repo (JaneliaSciComp/BigStitcher-Spark) is pure Java; its math
lives in the pinned Maven artifacts net.preibisch:BigStitcher:2.5.0 and
net.preibisch:multiview-reconstruction:8.0.0 (reference pom.xml:106-107),
whose sources are not vendored in /root/reference and cannot be fetched
(no network) or executed (no JVM/javac/mvn in this container — probed
2026-09-15). The reference's own test tree holds zero assertions, golden
vectors or fixtures, so there is nothing reference-side to
pin against. This package therefore restates the algorithm from (a) the
in-repo host semantics at the cited file:line, and (b) the published
algorithm of the pinned artifacts (Preibisch et al., Bioinformatics 2009;
Hörl et al., Nature Methods 2019), and pins behaviour itself via
known-answer property tests (tests/golden/) with analytically known
shifts and closed-form fused volumes. Every deliberate restatement choice
is marked [PIN] in the submodules.

Submodules:
  phasecorr — PairwiseStitching.getShift restatement
              (reference call site SparkPairwiseStitching.java:247-255)
  fusion    — BlkAffineFusion.initWithIntensityCoefficients restatement
              (reference call site SparkAffineFusion.java:602-615)
  synth     — seeded synthetic tile generator (BASELINE.json §(d) inputs)
"""

from . import phasecorr, fusion, synth  # noqa: F401
