"""sat_amd — an MI355X-native Show-Attend-and-Tell image-captioning framework.

Brand-new implementation (PyTorch-ROCm + hand-written CDNA4 HIP kernels + RCCL
over xGMI) of the capabilities of the TF-1.7 reference
Cheng-Lin-Li/show-attend-and-tell.  Layer map (mirrors SURVEY.md §1):

    CLI            main.py / main_distributed.py (repo root)
    Runtime        sat_amd.models.base_model.BaseModel
    Model          sat_amd.models.caption_generator.CaptionGenerator
    NN primitives  sat_amd.models.nn  (+ sat_amd.ops HIP kernel layer)
    Data & eval    sat_amd.data.*, sat_amd.evalcap.*
    Parallel       sat_amd.parallel.* (RCCL DP over xGMI)
"""

__version__ = "0.1.0"
