"""First-class component library (the reference's external ``detectmatelibrary``).

The ComponentResolver walks this package (resolver.py DEFAULT_ROOT) exactly
as the reference walks ``detectmatelibrary`` (component_resolver.py:75-93).
"""
