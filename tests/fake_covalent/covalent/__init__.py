"""Minimal covalent stand-in for integration tests: provides exactly the
four surfaces the plugin imports."""
