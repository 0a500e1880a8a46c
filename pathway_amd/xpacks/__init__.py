"""xpacks (reference python/pathway/xpacks): llm pipeline components."""
