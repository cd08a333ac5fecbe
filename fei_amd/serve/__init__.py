"""HTTP serving front-end (OpenAI-style) over the local MI355X engine."""
