"""Model zoo (reference: torchrec/models/)."""

from torchrec_amd.models.dlrm import (  # noqa: F401
    DLRM,
    DLRM_DCN,
    DLRM_Projection,
    DLRMTrain,
    DenseArch,
    InteractionArch,
    InteractionDCNArch,
    InteractionProjectionArch,
    OverArch,
    SparseArch,
)
from torchrec_amd.models.two_tower import (  # noqa: F401
    SequenceTwoTower,
    TwoTower,
    TwoTowerTrain,
)
from torchrec_amd.modules.deepfm import SimpleDeepFMNN  # noqa: F401
