print("hello from a substratus-amd notebook")
