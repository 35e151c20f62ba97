from deeprec_amd.models.dlrm import DLRM  # noqa: F401
