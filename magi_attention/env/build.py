"""Build-time env (reference env/build.py) - the MI355X engine builds
ahead of time via magi_attention.csrc.build."""
