from .rendezvous import ElasticContext, Scheduler  # noqa: F401
