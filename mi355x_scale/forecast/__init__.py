"""(placeholder — populated in later milestones)"""
