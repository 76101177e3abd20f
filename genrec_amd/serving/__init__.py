"""Serving package. Lazy exports so `python -m genrec_amd.serving.server`
doesn't double-import the module (runpy warning)."""


def __getattr__(name):
    if name in ("RecommendationService", "create_app"):
        from genrec_amd.serving import server

        return getattr(server, name)
    raise AttributeError(name)


__all__ = ["RecommendationService", "create_app"]
