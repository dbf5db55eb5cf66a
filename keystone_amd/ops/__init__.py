"""Operator library — the rebuild of KeystoneML's `nodes` packages
(SURVEY.md §2.2-§2.6): `learning` (solvers), `images`, `nlp`, `stats`,
`util`.  Import submodules directly, e.g.
``from keystone_amd.ops.learning import BlockLeastSquaresEstimator``."""

from . import images, learning, nlp, stats, util  # noqa: F401
