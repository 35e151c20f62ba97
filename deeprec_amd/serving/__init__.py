from deeprec_amd.serving.predictor import Predictor, SessionGroup
from deeprec_amd.serving.server import DynamicBatcher, create_app, serve

__all__ = ["Predictor", "SessionGroup", "DynamicBatcher", "create_app",
           "serve"]
