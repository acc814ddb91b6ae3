"""ai/ — the layout the reference README promises (README.md:72-76).
Thin shims over the nerrf_amd package."""
