from es_pytorch_amd.rollout.results import (TrainingResult, RewardResult, MeanRewardResult,  # noqa: F401
                                            DistResult, XDistResult, NSResult, NSRResult,
                                            MultiAgentTrainingResult)
from es_pytorch_amd.rollout.runner import run_model, multi_agent_runner  # noqa: F401
