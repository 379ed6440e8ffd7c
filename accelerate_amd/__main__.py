from .commands.cli import main

main()
