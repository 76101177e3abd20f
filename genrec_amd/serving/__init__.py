from genrec_amd.serving.server import RecommendationService, create_app

__all__ = ["RecommendationService", "create_app"]
