from .classifier import HashedTextClassifier, MODERATION_CATEGORIES, category_names  # noqa: F401
