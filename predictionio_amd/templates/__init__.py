"""Built-in MI355X-native engine templates (reference: examples/scala-parallel-*)."""
