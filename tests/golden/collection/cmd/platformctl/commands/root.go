/*
Copyright 2021.

Licensed under the Apache License, Version 2.0 (the "License");
you may not use this file except in compliance with the License.
You may obtain a copy of the License at

    http://www.apache.org/licenses/LICENSE-2.0

Unless required by applicable law or agreed to in writing, software
distributed under the License is distributed on an "AS IS" BASIS,
WITHOUT WARRANTIES OR CONDITIONS OF ANY KIND, either express or implied.
See the License for the specific language governing permissions and
limitations under the License.
*/

package commands

import (
	"github.com/spf13/cobra"

	// common imports for subcommands
	cmdgenerate "github.com/acme/platform/cmd/platformctl/commands/generate"
	cmdinit "github.com/acme/platform/cmd/platformctl/commands/init"
	cmdversion "github.com/acme/platform/cmd/platformctl/commands/version"

	// specific imports for workloads
	generateapps "github.com/acme/platform/cmd/platformctl/commands/generate/apps"
	generatedata "github.com/acme/platform/cmd/platformctl/commands/generate/data"
	generateplatforms "github.com/acme/platform/cmd/platformctl/commands/generate/platforms"
	initapps "github.com/acme/platform/cmd/platformctl/commands/init/apps"
	initdata "github.com/acme/platform/cmd/platformctl/commands/init/data"
	initplatforms "github.com/acme/platform/cmd/platformctl/commands/init/platforms"
	versionapps "github.com/acme/platform/cmd/platformctl/commands/version/apps"
	versiondata "github.com/acme/platform/cmd/platformctl/commands/version/data"
	versionplatforms "github.com/acme/platform/cmd/platformctl/commands/version/platforms"
	//+operator-builder:subcommands:imports
)

// PlatformctlCommand represents the base command when called without any subcommands.
type PlatformctlCommand struct {
	*cobra.Command
}

// NewPlatformctlCommand returns an instance of the PlatformctlCommand.
func NewPlatformctlCommand() *PlatformctlCommand {
	c := &PlatformctlCommand{
		Command: &cobra.Command{
			Use:   "platformctl",
			Short: "Manage the cloud platform",
			Long:  "Manage the cloud platform",
		},
	}

	c.addSubCommands()

	return c
}

// Run represents the main entry point into the command
// This is called by main.main() to execute the root command.
func (c *PlatformctlCommand) Run() {
	cobra.CheckErr(c.Execute())
}

func (c *PlatformctlCommand) newInitSubCommand() {
	parentCommand := cmdinit.GetParent(cmdinit.NewBaseInitSubCommand(c.Command))
	_ = parentCommand

	// add the init subcommands
	initplatforms.NewCloudPlatformSubCommand(parentCommand)
	initapps.NewWebAppSubCommand(parentCommand)
	initdata.NewDataStoreSubCommand(parentCommand)
	//+operator-builder:subcommands:init
}

func (c *PlatformctlCommand) newGenerateSubCommand() {
	parentCommand := cmdgenerate.GetParent(cmdgenerate.NewBaseGenerateSubCommand(c.Command))
	_ = parentCommand

	// add the generate subcommands
	generateplatforms.NewCloudPlatformSubCommand(parentCommand)
	generateapps.NewWebAppSubCommand(parentCommand)
	generatedata.NewDataStoreSubCommand(parentCommand)
	//+operator-builder:subcommands:generate
}

func (c *PlatformctlCommand) newVersionSubCommand() {
	parentCommand := cmdversion.GetParent(cmdversion.NewBaseVersionSubCommand(c.Command))
	_ = parentCommand

	// add the version subcommands
	versionplatforms.NewCloudPlatformSubCommand(parentCommand)
	versionapps.NewWebAppSubCommand(parentCommand)
	versiondata.NewDataStoreSubCommand(parentCommand)
	//+operator-builder:subcommands:version
}

// addSubCommands adds any additional subCommands to the root command.
func (c *PlatformctlCommand) addSubCommands() {
	c.newInitSubCommand()
	c.newGenerateSubCommand()
	c.newVersionSubCommand()
}
